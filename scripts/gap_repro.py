import sys, torch, numpy as np
sys.path.insert(0, '/root/repo')
from bench import make_synthetic_windows
from deepconsensus_amd.models import config as cfg
from deepconsensus_amd.models.model import get_model
from deepconsensus_amd.models.runner import InferenceRunner
params = cfg.get_config("transformer_learn_values+custom")
cfg.modify_params(params, is_training=False)
torch.manual_seed(1234)
runner = InferenceRunner(params, get_model(params), device="cuda")
print("native:", runner.native)
rows = make_synthetic_windows(params, 4096, 7)
for b in (64, 1908, 4096):
    x = torch.from_numpy(rows[:b].astype(np.int16))
    bases, quals = runner.forward_windows(x)
    g = (bases == 0).float().mean().item()
    print(f"B={b}: gap_frac={g:.4f} uniq={bases.unique().tolist()[:6]}")
