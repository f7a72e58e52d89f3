import random, sys, os
sys.path.insert(0, "/root/repo")
import numpy as np, torch
torch.cuda.init()
from baikaldb_amd import GpuEngine
from oracle import Oracle
import tests.test_gpu_fuzz as fz
from tests.test_gpu_agg import run_both, assert_parity
eng, orc = GpuEngine(), Oracle()
lo, hi = int(sys.argv[1]), int(sys.argv[2])
for cs in range(lo, hi):
    rng = random.Random(90_000 + cs)
    specs, conjuncts, group, aggs = fz.random_case(rng)
    n = rng.choice([1000, 20_000, 120_000])
    print(f"case {cs}: n={n} specs={specs} group={group}", flush=True)
    got, exp = run_both(eng, orc, specs, n, conjuncts, group, aggs,
                        seed=rng.randrange(1 << 40), expected_groups=1 << 12)
    assert_parity(got, exp, aggs, [s[0] for s in specs])
print("RANGE OK", flush=True)
