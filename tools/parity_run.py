"""Loss-parity artifact: tiny-llama 30 fixed steps; run on GPU (bf16) and
CPU (fp32); losses written to JSON for the parity report."""
import json, os, sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from vescale_amd.fsdp import FSDP, FlatAdamW
from vescale_amd.models.llama import LlamaModel, llama_tiny

dev = torch.device("cuda" if torch.cuda.is_available() else "cpu")
dtype = torch.bfloat16 if dev.type == "cuda" else torch.float32
torch.manual_seed(42)
cfg = llama_tiny()
model = LlamaModel(cfg).to(dev)
model.init_weights()
eng = FSDP(model, None, param_dtype=dtype, device=dev)
opt = FlatAdamW(eng, lr=1e-3, grad_clip=1.0, weight_decay=0.0)
g = torch.Generator().manual_seed(1234)
losses = []
for step in range(30):
    x = torch.randint(0, cfg.vocab_size, (4, 64), generator=g).to(dev)
    y = torch.roll(x, -1, dims=1)
    loss = eng(x, y)
    loss.backward()
    opt.step()
    losses.append(float(loss))
out = sys.argv[1] if len(sys.argv) > 1 else "losses.json"
json.dump({"device": dev.type, "dtype": str(dtype), "losses": losses}, open(out, "w"))
print("final loss", losses[-1])
