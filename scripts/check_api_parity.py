#!/usr/bin/env python3
"""Module-level public-name parity vs the reference package.

For every mapped module, lists reference public names (top-level
functions/classes/assignments not starting with '_') missing from the
brainiak_amd counterpart.  Zero output lines (beyond the summary)
means full surface parity.

Usage: python scripts/check_api_parity.py [/root/reference/src/brainiak]
"""
import ast
import pathlib
import sys

MAP = {
    'fcma/preprocessing.py': 'fcma/preprocessing.py',
    'fcma/voxelselector.py': 'fcma/voxelselector.py',
    'fcma/classifier.py': 'fcma/classifier.py',
    'fcma/mvpa_voxelselector.py': 'fcma/mvpa_voxelselector.py',
    'fcma/util.py': 'fcma/util.py',
    'searchlight/searchlight.py': 'searchlight/searchlight.py',
    'funcalign/srm.py': 'funcalign/srm.py',
    'funcalign/rsrm.py': 'funcalign/rsrm.py',
    'funcalign/sssrm.py': 'funcalign/sssrm.py',
    'funcalign/fastsrm.py': 'funcalign/fastsrm.py',
    'eventseg/event.py': 'eventseg/event.py',
    'factoranalysis/tfa.py': 'factoranalysis/tfa.py',
    'factoranalysis/htfa.py': 'factoranalysis/htfa.py',
    'isc.py': 'isc.py',
    'reprsimil/brsa.py': 'reprsimil/brsa.py',
    'matnormal/covs.py': 'matnormal/covs.py',
    'matnormal/regression.py': 'matnormal/regression.py',
    'matnormal/mnrsa.py': 'matnormal/mnrsa.py',
    'matnormal/utils.py': 'matnormal/utils.py',
    'matnormal/matnormal_likelihoods.py':
        'matnormal/matnormal_likelihoods.py',
    'reconstruct/iem.py': 'reconstruct/iem.py',
    'hyperparamopt/hpo.py': 'hyperparamopt/hpo.py',
    'utils/utils.py': 'utils/utils.py',
    'utils/fmrisim.py': 'utils/fmrisim.py',
    'utils/fmrisim_real_time_generator.py':
        'utils/fmrisim_real_time_generator.py',
    'utils/kronecker_solvers.py': 'utils/kronecker_solvers.py',
    'image.py': 'image.py',
    'io.py': 'io.py',
}


# deliberate renames (no TensorFlow in this framework): the check
# passes when the rename target exists in the counterpart module
RENAMES = {
    'tf_kron_mult': 'kron_mult',
    'tf_masked_triangular_solve': 'masked_triangular_solve',
    'tf_solve_lower_triangular_kron': 'solve_lower_triangular_kron',
    'tf_solve_lower_triangular_masked_kron':
        'solve_lower_triangular_masked_kron',
    'tf_solve_upper_triangular_kron': 'solve_upper_triangular_kron',
    'tf_solve_upper_triangular_masked_kron':
        'solve_upper_triangular_masked_kron',
}


def public_names(path):
    tree = ast.parse(path.read_text())
    names = set()
    for node in tree.body:
        if isinstance(node, (ast.FunctionDef, ast.AsyncFunctionDef,
                             ast.ClassDef)):
            if not node.name.startswith('_'):
                names.add(node.name)
        elif isinstance(node, ast.Assign):
            for t in node.targets:
                if isinstance(t, ast.Name) and not t.id.startswith('_'):
                    names.add(t.id)
    names.discard('logger')
    return names


def main():
    ref_root = pathlib.Path(sys.argv[1] if len(sys.argv) > 1
                            else '/root/reference/src/brainiak')
    mine_root = pathlib.Path(__file__).resolve().parent.parent \
        / 'brainiak_amd'
    missing_total = 0
    for ref_rel, my_rel in sorted(MAP.items()):
        ref_p = ref_root / ref_rel
        my_p = mine_root / my_rel
        if not ref_p.exists():
            print(f"[skip] reference missing {ref_rel}")
            continue
        mine = public_names(my_p)
        missing = {n for n in public_names(ref_p) - mine
                   if RENAMES.get(n) not in mine}
        for name in sorted(missing):
            print(f"{my_rel}: missing public name '{name}'")
        missing_total += len(missing)
    print(f"-- {missing_total} missing public names across "
          f"{len(MAP)} modules")
    return 1 if missing_total else 0


if __name__ == '__main__':
    raise SystemExit(main())
