"""Roofline evidence for the hipBLASLt share of the BERT/GPT-2 step
(VERDICT r1 item 7): measure every hot GEMM shape (fwd + dgrad + wgrad
layouts exactly as torch autograd issues them) and report achieved
TFLOP/s vs the 2.5 PF/s dense bf16 MFMA peak."""
import os, sys, time
sys.path.insert(0, "/root/repo")
import torch

# load the TunableOp table the bench uses
os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "0")
tdir = "/root/repo/profiles/tunableop"
if os.path.isdir(tdir):
    os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME",
                          os.path.join(tdir, "tunableop_results%d.csv"))

PEAK = 2500.0  # TF/s bf16 dense (no sparsity)


def bench(fn, iters=20, warm=5):
    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def gemm_suite(tag, m, k, n):
    """One Linear(in=k, out=n) at batch m: fwd y=x@W^T, dgrad dx=dy@W,
    wgrad dW=dy^T@x — the exact layouts torch's linear backward hits."""
    x = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(n, k, device="cuda", dtype=torch.bfloat16)
    dy = torch.randn(m, n, device="cuda", dtype=torch.bfloat16)
    fl = 2.0 * m * k * n
    rows = []
    t = bench(lambda: torch.nn.functional.linear(x, w))
    rows.append(("fwd  (NT)", t))
    t = bench(lambda: dy @ w)
    rows.append(("dgrad(NN)", t))
    t = bench(lambda: dy.t() @ x)
    rows.append(("wgrad(TN)", t))
    for name, t in rows:
        tf = fl / t / 1e12
        print(f"{tag:28s} {name}  m{m} k{k} n{n}: {t*1e6:8.1f} us "
              f"{tf:7.0f} TF  {100*tf/PEAK:4.1f}% of peak")
    return [(fl, t) for _, t in rows]


print("== BERT-Large b128 s512 (tokens 65536) ==")
allr = []
allr += gemm_suite("bert qkv", 65536, 1024, 3072)
allr += gemm_suite("bert proj", 65536, 1024, 1024)
allr += gemm_suite("bert fc1", 65536, 1024, 4096)
allr += gemm_suite("bert fc2 ", 65536, 4096, 1024)
print("== GPT-2 XL b16 s1024 (tokens 16384) ==")
allr += gemm_suite("gpt2 qkv", 16384, 1600, 4800)
allr += gemm_suite("gpt2 proj", 16384, 1600, 1600)
allr += gemm_suite("gpt2 fc1", 16384, 1600, 6400)
allr += gemm_suite("gpt2 fc2", 16384, 6400, 1600)
print("== vocab projections ==")
allr += gemm_suite("bert mlm head", 65536, 1024, 30522)
allr += gemm_suite("gpt2 lm head", 16384, 1600, 50257)

tot_fl = sum(f for f, _ in allr)
tot_t = sum(t for _, t in allr)
print(f"\nAGGREGATE over all hot GEMM shapes: {tot_fl/tot_t/1e12:.0f} TF "
      f"= {100*tot_fl/tot_t/1e12/PEAK:.1f}% of 2.5 PF dense bf16 peak")


# wgrad-TN probe: is the slow TN layout hipBLASLt's kernel choice, or
# would an explicit transpose-copy + NN GEMM beat it?
print("\n== wgrad TN vs copy+NN probe ==")
for (tag, m, k, n) in (("bert proj", 65536, 1024, 1024),
                       ("bert qkv", 65536, 1024, 3072),
                       ("gpt2 fc2", 16384, 6400, 1600)):
    x = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
    dy = torch.randn(m, n, device="cuda", dtype=torch.bfloat16)
    fl = 2.0 * m * k * n
    t_tn = bench(lambda: dy.t() @ x)
    t_cp = bench(lambda: dy.t().contiguous() @ x)
    print(f"{tag:12s} TN {fl/t_tn/1e12:6.0f} TF | transpose-copy+NN "
          f"{fl/t_cp/1e12:6.0f} TF")
