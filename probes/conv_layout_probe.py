import torch, time, os
torch.backends.cudnn.benchmark = True
dev = "cuda"
def bench_conv(dtype, channels_last, label, C=3, chans=(32,64,128,256)):
    convs = []
    inc = C
    for oc in chans:
        c = torch.nn.Conv2d(inc, oc, 4, 2, 1, bias=False).to(dev, dtype)
        if channels_last: c = c.to(memory_format=torch.channels_last)
        convs.append(c); inc = oc
    x = torch.randn(1024, C, 64, 64, device=dev, dtype=dtype, requires_grad=True)
    if channels_last: x = x.to(memory_format=torch.channels_last).detach().requires_grad_()
    def step():
        y = x
        for c in convs: y = c(y)
        y.sum().backward()
        x.grad = None
        for c in convs: c.weight.grad = None
    for _ in range(3): step()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(10): step()
    torch.cuda.synchronize()
    print(f"{label:40s} {(time.perf_counter()-t0)/10*1000:8.2f} ms/iter")

def bench_deconv(dtype, channels_last, label):
    convs = []
    chans = [(256,128),(128,64),(64,32),(32,3)]
    mods = []
    for ic,oc in chans:
        c = torch.nn.ConvTranspose2d(ic, oc, 4, 2, 1, bias=False).to(dev, dtype)
        if channels_last: c = c.to(memory_format=torch.channels_last)
        mods.append(c)
    x = torch.randn(1024, 256, 4, 4, device=dev, dtype=dtype, requires_grad=True)
    if channels_last: x = x.to(memory_format=torch.channels_last).detach().requires_grad_()
    def step():
        y = x
        for c in mods: y = c(y)
        y.sum().backward()
        x.grad = None
        for c in mods: c.weight.grad = None
    for _ in range(3): step()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(10): step()
    torch.cuda.synchronize()
    print(f"{label:40s} {(time.perf_counter()-t0)/10*1000:8.2f} ms/iter")

print("MIOPEN_FIND_MODE =", os.environ.get("MIOPEN_FIND_MODE"))
bench_conv(torch.bfloat16, False, "conv bf16 NCHW")
bench_conv(torch.bfloat16, True,  "conv bf16 channels_last")
bench_conv(torch.float32, False,  "conv fp32 NCHW")
bench_conv(torch.float32, True,   "conv fp32 channels_last")
bench_deconv(torch.bfloat16, False, "deconv bf16 NCHW")
bench_deconv(torch.bfloat16, True,  "deconv bf16 channels_last")
bench_deconv(torch.float32, False,  "deconv fp32 NCHW")
bench_deconv(torch.float32, True,   "deconv fp32 channels_last")
