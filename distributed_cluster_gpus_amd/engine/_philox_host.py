"""Host-side Philox4x32-10 reference, bit-identical to the device
implementation (ops/csrc/hip/philox.hpp).  Used to seed initial arrival
times consistently with the device streams, and by tests as the numerics
reference for the GPU RNG."""

M0 = 0xD2511F53
M1 = 0xCD9E8D57
B0 = 0x9E3779B9
B1 = 0xBB67AE85
MASK = 0xFFFFFFFF


def philox4x32(key: int, ctr: int):
    c0 = ctr & MASK
    c1 = (ctr >> 32) & MASK
    c2, c3 = 0, 0
    k0 = key & MASK
    k1 = (key >> 32) & MASK
    for _ in range(10):
        p0 = (M0 * c0) & 0xFFFFFFFFFFFFFFFF
        p1 = (M1 * c2) & 0xFFFFFFFFFFFFFFFF
        h0, l0 = (p0 >> 32) & MASK, p0 & MASK
        h1, l1 = (p1 >> 32) & MASK, p1 & MASK
        c0, c1, c2, c3 = (h1 ^ c1 ^ k0) & MASK, l1, (h0 ^ c3 ^ k1) & MASK, l0
        k0 = (k0 + B0) & MASK
        k1 = (k1 + B1) & MASK
    return c0, c1, c2, c3


def philox_u01(key: int, ctr: int) -> float:
    w = philox4x32(key, ctr)
    a, b = w[0] >> 5, w[1] >> 6
    return (a * 67108864.0 + b) * (1.0 / 9007199254740992.0)


def philox_u01_pair(key: int, ctr: int):
    w = philox4x32(key, ctr)
    ua = ((w[0] >> 5) * 67108864.0 + (w[1] >> 6)) * (1.0 / 9007199254740992.0)
    ub = ((w[2] >> 5) * 67108864.0 + (w[3] >> 6)) * (1.0 / 9007199254740992.0)
    return ua, ub


def replica_key(seed: int, global_replica_id: int) -> int:
    return (seed ^ ((0x9E3779B97F4A7C15 * global_replica_id) & 0xFFFFFFFFFFFFFFFF)) \
        & 0xFFFFFFFFFFFFFFFF
