import sys

import torch
sys.path.insert(0, "/root/repo")
from zero_transformer_amd.utils import gemm_tune
assert gemm_tune.enable(tuning=True)
torch.cuda.tunable.set_max_tuning_duration(50)
from zero_transformer_amd.models.inference import model_getter, generate_fast
dev = torch.device("cuda", 0)
model = model_getter("1_3b").to(dev).half().eval()
for b in (1, 16):
    idx = torch.randint(0, model.vocab_size, (b, 64), device=dev)
    generate_fast(model, idx, 4, use_graph=False)  # tuning hates capture
torch.cuda.synchronize()
print("tuned decode shapes")
