import sys, torch
sys.path.insert(0, "/root/repo")
import msbn

case = sys.argv[1]
dev = "cuda:0"
torch.manual_seed(21)

if case == "a":   # test config exactly
    model = msbn.models.resnet18(fused=True).to(dev); model.train()
    x = torch.randn(8, 3, 64, 64, device=dev)
elif case == "b": # bench-like config (bf16 CL resnet18 fused)
    from bench import cast_bf16_keep_bn_fp32
    model = cast_bf16_keep_bn_fp32(msbn.models.resnet18(fused=True).to(dev))
    model = model.to(memory_format=torch.channels_last); model.train()
    x = torch.randn(8, 3, 64, 64, device=dev, dtype=torch.bfloat16).to(
        memory_format=torch.channels_last)
elif case == "c": # test config, unfused model
    model = msbn.models.resnet18(fused=False).to(dev); model.train()
    x = torch.randn(8, 3, 64, 64, device=dev)
elif case == "d": # fwd-only capture of fused fp32
    model = msbn.models.resnet18(fused=True).to(dev); model.train()
    x = torch.randn(8, 3, 64, 64, device=dev)

y = torch.randint(0, 1000, (8,), device=dev)
opt = torch.optim.SGD(model.parameters(), lr=0.01, momentum=0.9)

def step():
    opt.zero_grad(set_to_none=False)
    loss = torch.nn.functional.cross_entropy(model(x).float(), y)
    if case != "d":
        loss.backward(); opt.step()
    return loss

for _ in range(3):
    step()
torch.cuda.synchronize()
g = torch.cuda.CUDAGraph()
with torch.cuda.graph(g):
    step()
w0 = None
if case != "d":
    w0 = model.fc.weight.detach().clone()
for _ in range(2):
    g.replay()
torch.cuda.synchronize()
if w0 is not None:
    assert not torch.equal(model.fc.weight, w0), "params must move on replay"
print("CASE", case, "OK")
