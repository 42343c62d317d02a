# Compare the native NHWC ResNet-18 against the fp32 torch twin.
import torch
import torch.nn.functional as F

from flashy_amd.models import native_resnet18, resnet18

torch.manual_seed(5)
twin = resnet18(num_classes=10, small_input=True).cuda().train()
model = native_resnet18(10).cuda().train().from_torch(twin)
x = torch.randn(8, 3, 32, 32, device="cuda")
y = torch.randint(10, (8,), device="cuda")
logits_n = model(x)
logits_t = twin(x)
ln = F.cross_entropy(logits_n, y)
lt = F.cross_entropy(logits_t, y)
print("loss", ln.item(), lt.item())
print("max diff", (logits_n - logits_t).abs().max().item(),
      "std", logits_t.std().item())
ln.backward()
lt.backward()
gn = model.layer1[0].conv1.weight.grad.permute(0, 3, 1, 2).flatten()
gt = twin.layer1[0].conv1.weight.grad.flatten()
print("grad relL2", ((gn - gt).norm() / (gt.norm() + 1e-8)).item(),
      "cos", torch.nn.functional.cosine_similarity(gn, gt, dim=0).item())

# per-stage comparison to localize drift
with torch.no_grad():
    xb = x.permute(0, 2, 3, 1).contiguous().to(torch.bfloat16)
    a_n = model.stem_bn(model.stem_conv(xb), relu=True)
    a_t = twin.stem(x)
    d = (a_n.float().permute(0, 3, 1, 2) - a_t).abs().max().item()
    print("after stem:", d, a_t.abs().max().item())
    b_n = model.layer1(a_n)
    b_t = twin.layer1(a_t)
    d = (b_n.float().permute(0, 3, 1, 2) - b_t).abs().max().item()
    print("after layer1:", d, b_t.abs().max().item())
