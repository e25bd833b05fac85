import faulthandler, sys
faulthandler.enable()
import tempfile
import numpy as np
sys.path.insert(0, '/root/repo')
from paimon_amd.datagen import gen_runs_c5, write_runs_c5, C5_VALUE_COLS
from paimon_amd import Session, MergeReadPlan, file_descs_from_metas
d = tempfile.mkdtemp()
runs = gen_runs_c5(2, 3000, seed=61, str_card=50)
metas = write_runs_c5(runs, d)
print('wrote', flush=True)
s = Session(0)
print('session', flush=True)
plan = MergeReadPlan(s, file_descs_from_metas(metas),
                     [{"name": "_KEY_k", "type": "int64"}], C5_VALUE_COLS)
print('plan created', flush=True)
b = plan.read_next()
print('read', None if b is None else len(b.get('_KEY_k', [])), flush=True)
print({k: (v[:5] if hasattr(v, '__len__') else v) for k, v in b.items()},
      flush=True)
plan.close(); s.close()
print('done', flush=True)
