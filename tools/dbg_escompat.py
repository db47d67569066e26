import json, sys, os
REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
os.chdir(REPO)
sys.path.insert(0, os.path.join(REPO, "tests"))
sys.path.insert(0, REPO)
from fastapi.testclient import TestClient
from quickwit_amd.rest import create_app
from quickwit_amd.api import GpuSearcher
from rest_replay import run_step
from test_rest_scenarios import skip_step
import sys as _s
suite = _s.argv[1] if len(_s.argv) > 1 else 'es_compatibility'
steps = json.load(open('tests/golden/rest_scenarios.json'))['suites'][suite]
c = TestClient(create_app(lambda: GpuSearcher(device=0)))
for i, s in enumerate(steps):
    if skip_step(i, s):
        continue
    try:
        run_step(c, s)
    except Exception as e:
        print('FAIL', i, json.dumps(s.get('json', s.get('params', {})))[:160])
        print('  ', str(e)[:250])
        r = c.request(s['method'], '/api/v1/' + s['endpoint'], json=s.get('json'))
        print('   got:', json.dumps(r.json())[:400])
print('done')
