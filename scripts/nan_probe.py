import sys, torch
sys.path.insert(0, ".")
from dnet_amd.models import ModelConfig, PRESETS
from dnet_amd.parallel.ring import RingExecutor
import dnet_amd.models.moe as moe

cfg = ModelConfig.from_hf(dict(PRESETS["gpt-oss-20b"]), quant=None)
ex = RingExecutor(cfg, 0, 1, torch.device("cuda:0"), mb_count=1, mb_size=32,
                  smax=256, seed=1234, use_graphs=False)
orig = moe.stack_route_weights
stats = []
def spy(st, we, tp, tr):
    nz = int((we != 0).sum())
    nan = int(torch.isnan(we).sum())
    tiles = []
    for m0 in range(0, we.shape[0], 8):
        tiles.append(int(((we[m0:m0+8] != 0).any(0)).sum()))
    stats.append((nz, nan, tiles))
    return orig(st, we, tp, tr)
moe.stack_route_weights = spy
toks = torch.randint(100, 200, (1, 32, 16))
ex.prefill(toks)
stats.clear()
ex.decode_rounds(1)
print("layers seen:", len(stats))
for s in stats[:4]:
    print("nonzero", s[0], "nan", s[1], "experts-per-8row-tile", s[2])
