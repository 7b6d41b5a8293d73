"""Randomized on-GPU fuzz of the gfx950 kernels against torch references:
random region sizes, random dirty-page patterns (none/all/adjacent/sparse)
through DeviceSnapshot diff+apply, and random elementwise op/dtype/size
combos. Run: python tools/gpu_fuzz.py [rounds]
"""

import random
import sys

sys.path.insert(0, __file__.rsplit("/", 2)[0])

import torch  # noqa: E402

from faabric_amd import _core  # noqa: E402

OPS = {
    0: lambda a, b: a + b,
    1: torch.maximum,
    2: torch.minimum,
    3: lambda a, b: a * b,
}
DTYPES = {0: torch.int32, 3: torch.float32, 4: torch.float64}


def fuzz_snapshot(rng):
    n_pages = rng.choice([1, 2, 3, 7, 64, 257, 1024])
    n = n_pages * 4096
    base = torch.randint(0, 256, (n,), dtype=torch.uint8, device="cuda")
    snap = _core.DeviceSnapshot(n, 0)
    torch.cuda.synchronize()
    snap.capture_from_ptr(base.data_ptr())

    mode = rng.choice(["none", "all", "adjacent", "sparse", "single-byte"])
    updated = base.clone()
    if mode == "all":
        updated ^= 0xFF
    elif mode == "adjacent":
        s = rng.randrange(n_pages)
        e = min(n_pages, s + rng.randrange(1, 4))
        updated[s * 4096 : e * 4096] ^= 0x3C
    elif mode == "sparse":
        for p in rng.sample(range(n_pages), min(n_pages, 5)):
            updated[p * 4096 + rng.randrange(4096)] ^= 0x01
    elif mode == "single-byte":
        updated[rng.randrange(n)] ^= 0x80
    torch.cuda.synchronize()

    expected_dirty = int(
        (updated.view(n_pages, 4096) != base.view(n_pages, 4096))
        .any(dim=1)
        .sum()
    )
    nd = snap.diff_xor(updated.data_ptr())
    assert nd == expected_dirty, (mode, nd, expected_dirty)
    if nd > 0:
        snap.apply_last_diff()
    got = snap.copy_out_host(n)
    assert got == updated.cpu().numpy().tobytes(), mode


def fuzz_elementwise(rng):
    dtype = rng.choice(list(DTYPES))
    op = rng.choice(list(OPS))
    n = rng.choice([1, 63, 64, 65, 1000, 12345, 1 << 18, (1 << 18) + 7])
    td = DTYPES[dtype]
    if td == torch.int32:
        a = torch.randint(-9999, 9999, (n,), dtype=td, device="cuda")
        b = torch.randint(-9999, 9999, (n,), dtype=td, device="cuda")
    else:
        a = torch.randn(n, dtype=td, device="cuda")
        b = torch.randn(n, dtype=td, device="cuda")
    ref = OPS[op](a.clone(), b)
    torch.cuda.synchronize()
    _core.device_elementwise_op(a.data_ptr(), b.data_ptr(), n, dtype, op)
    torch.cuda.synchronize()
    assert torch.equal(a, ref), (dtype, op, n)


def main(rounds=60):
    assert torch.cuda.is_available()
    rng = random.Random(20260913)
    for i in range(rounds):
        fuzz_snapshot(rng)
        fuzz_elementwise(rng)
        if (i + 1) % 20 == 0:
            print(f"round {i+1}/{rounds} ok", flush=True)
    print(f"GPU FUZZ OK: {rounds} rounds")


if __name__ == "__main__":
    main(int(sys.argv[1]) if len(sys.argv) > 1 else 60)
