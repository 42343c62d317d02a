# In-process restore check: solver1 trains+commits (async or sync per argv),
# solver2 restores from the file and continues; their next-10-step losses
# should be in the same regime.
import os, sys
sys.path.insert(0, ".")
os.environ["_FLASHY_AMD_DIR"] = "/tmp/arc_xp"
import torch
from flashy_amd import xp as fxp
from flashy_amd.config import Config
from flashy_amd.models import native_resnet18
from flashy_amd.optim import FusedSGD
from flashy_amd.functional import cross_entropy
from flashy_amd.solver import BaseSolver

ASYNC = sys.argv[1] == "async"
torch.manual_seed(0)
X = torch.randn(64, 3, 32, 32, device="cuda")
Y = torch.randint(10, (64,), device="cuda")

class S(BaseSolver):
    def __init__(self):
        super().__init__()
        self.async_checkpoint = ASYNC
        torch.manual_seed(7)
        self.model = native_resnet18(10).cuda().train()
        self.optim = FusedSGD(self.model.parameters(), lr=0.1, momentum=0.9,
                              weight_decay=5e-4, bf16_mirror=True)
        self.model.enable_wt_cache()
        self.register_stateful("model", "optim")
    def steps(self, n):
        losses = []
        for _ in range(n):
            loss = cross_entropy(self.model(X), Y)
            self.optim.zero_grad()
            loss.backward()
            self.optim.step()
            losses.append(round(loss.item(), 3))
        return losses
    def run(self):
        for _ in range(2):
            self.run_stage("train", lambda: {"l": self.steps(20)[-1]})
            self.commit()
        self.finalize_checkpoint()

fxp.create_xp(Config.wrap({"tag": 1})).enter()
s1 = S()
s1.run()
cont = s1.steps(10)
print("live continue :", cont[:5])

fxp._current_xp = None
fxp.create_xp(Config.wrap({"tag": 1})).enter()
s2 = S()
assert s2.restore()
s2.optim.refresh_bf16()
res = s2.steps(10)
print("after restore :", res[:5])
