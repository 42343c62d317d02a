# In-process check: after async commits, the checkpoint file must equal the
# live model/optimizer state exactly.
import os, sys, pathlib
sys.path.insert(0, ".")
os.environ["_FLASHY_AMD_DIR"] = "/tmp/ack_xp"
import torch
from flashy_amd import xp as fxp
from flashy_amd.config import Config
from flashy_amd import checkpoint as fckpt
from flashy_amd.models import native_resnet18
from flashy_amd.optim import FusedSGD
from flashy_amd.functional import cross_entropy
from flashy_amd.solver import BaseSolver

class S(BaseSolver):
    async_checkpoint = True
    def __init__(self):
        super().__init__()
        torch.manual_seed(0)
        self.model = native_resnet18(10).cuda().train()
        self.optim = FusedSGD(self.model.parameters(), lr=0.05, momentum=0.9,
                              bf16_mirror=True)
        self.model.enable_wt_cache()
        self.register_stateful("model", "optim")
    def train_one(self):
        x = torch.randn(32, 3, 32, 32, device="cuda")
        y = torch.randint(10, (32,), device="cuda")
        for _ in range(10):
            loss = cross_entropy(self.model(x), y)
            self.optim.zero_grad()
            loss.backward()
            self.optim.step()
        return {"loss": loss.item()}
    def run(self):
        for _ in range(2):
            self.run_stage("train", self.train_one)
            self.commit()
        self.finalize_checkpoint()

fxp.create_xp(Config.wrap({"lr": 0.05})).enter()
s = S()
s.run()
state = fckpt.load_state(s.checkpoint_path)
live = s.model.state_dict()
bad = 0
for k, v in live.items():
    got = state["model"][k]
    if not torch.equal(got.cpu(), v.detach().cpu()):
        d = (got.cpu().float() - v.detach().cpu().float()).abs().max().item()
        print("MISMATCH", k, "maxdiff", d)
        bad += 1
        if bad > 5: break
mom_live = s.optim._momentum_buffers[0].cpu()
mom_ck = state["optim"]["extra"]["momentum_buffers"][0].cpu()
print("momentum equal:", torch.equal(mom_live, mom_ck),
      "maxdiff", (mom_live - mom_ck).abs().max().item())
print("OK" if bad == 0 else f"{bad}+ tensor mismatches")
