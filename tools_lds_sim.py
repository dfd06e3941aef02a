"""LDS bank-conflict simulator for the attention kernels (gfx950 model from
/opt/skills/guides/MI355X_MICROARCH.md §LDS):

  ds_read_b128 : 4 lane groups {0-3,12-15,20-27},{4-11,16-19,28-31},
                 {32-35,44-47,52-59},{36-43,48-51,60-63}; bank=(a/4)%64;
                 each lane touches 4 consecutive banks; within a group each
                 extra distinct address on a busy bank adds one cycle.
  ds_write_b128: 8 groups of 8 contiguous lanes; bank=(a/4)%32.
  b16/b32 ops  : 2 groups of 32 contiguous lanes; bank=(a/4)%32 (1 dword).

Reports cycles vs ideal for each named access in fa_fwd / fa_bwd_dkv /
fa_bwd_dq. Pure analysis tool; not part of the product or tests."""

from collections import defaultdict

G128_READ = [
    [0, 1, 2, 3, 12, 13, 14, 15, 20, 21, 22, 23, 24, 25, 26, 27],
    [4, 5, 6, 7, 8, 9, 10, 11, 16, 17, 18, 19, 28, 29, 30, 31],
    [32, 33, 34, 35, 44, 45, 46, 47, 52, 53, 54, 55, 56, 57, 58, 59],
    [36, 37, 38, 39, 40, 41, 42, 43, 48, 49, 50, 51, 60, 61, 62, 63],
]
G128_WRITE = [list(range(i * 8, i * 8 + 8)) for i in range(8)]
G32 = [list(range(0, 32)), list(range(32, 64))]


def cycles(addrs_bytes, groups, ndwords, bankmod):
    """addrs_bytes: lane -> byte address (or None for inactive). Returns
    (cycles, ideal_cycles)."""
    total = 0
    for grp in groups:
        # bank -> set of distinct dword addresses
        banks = defaultdict(set)
        for lane in grp:
            a = addrs_bytes.get(lane)
            if a is None:
                continue
            for d in range(ndwords):
                dw = a // 4 + d
                banks[dw % bankmod].add(dw)
            # one LDS cycle minimum per group that has any active lane
        if banks:
            total += max(len(v) for v in banks.values())
    ideal = sum(1 for grp in groups if any(addrs_bytes.get(l) is not None for l in grp))
    return total, ideal


def read128(addrs):
    return cycles(addrs, G128_READ, 4, 64)


def write128(addrs):
    return cycles(addrs, G128_WRITE, 4, 32)


def write16(addrs):
    return cycles(addrs, G32, 1, 32)


def report(name, pairs):
    c = sum(p[0] for p in pairs)
    i = sum(p[1] for p in pairs)
    print(f"  {name:42s} cycles={c:5d} ideal={i:5d} factor={c / max(i, 1):.2f}")
    return c, i


def SWZ(row, col, ST):
    return row * ST + ((((col >> 3) ^ ((row >> 3) & 7)) << 3) | (col & 7))


def lane_split(l):
    return l & 15, l >> 4  # lr, lg


def main():
    DPAD = 96
    ST = 72
    SQ = DPAD + 8  # 104
    SK = DPAD + 8
    SV = 72

    print("== fa_bwd_dkv ==")
    tot = [0, 0]

    def acc(name, pairs):
        c, i = report(name, pairs)
        tot[0] += c
        tot[1] += i

    # staging: thread t handles pieces pidx = t and t+512 (768 pieces)
    # b128 writes Qlds[qq*SQ + d0]
    for base in (0, 512):
        addrs = {}
        addrsT = [dict() for _ in range(8)]
        for l in range(64):
            # wave 0 lanes are threads 0..63 (+base offset handled per wave;
            # conflicts are per wave-instruction — model wave 0's instr)
            pidx = base + l
            if pidx >= 768:
                continue
            qq = pidx // (DPAD // 8)
            d0 = (pidx % (DPAD // 8)) * 8
            addrs[l] = (qq * SQ + d0) * 2
            for e in range(8):
                addrsT[e][l] = SWZ(d0 + e, qq, ST) * 2
        acc(f"stage b128 write Qlds (piece base {base})", [write128(addrs)])
        acc(f"stage 8x b16 scatter QTl (base {base})", [write16(a) for a in addrsT])

    # compute reads: Qlds[(cb*16+lr)*SQ + kc*32+lg*8] b128
    pairs = []
    for cb in range(4):
        for kc in range(DPAD // 32):
            addrs = {}
            for l in range(64):
                lr, lg = lane_split(l)
                addrs[l] = ((cb * 16 + lr) * SQ + kc * 32 + lg * 8) * 2
            pairs.append(read128(addrs))
    acc("S^T/dP^T B-frag reads Qlds (x2 for dOl)", pairs)

    # P^T/dS^T strip writes: PTl[(wave*16+lg*4+r)*ST + cb*16+lr] b16 (wave 0)
    pairs = []
    for cb in range(4):
        for r in range(4):
            addrs = {}
            for l in range(64):
                lr, lg = lane_split(l)
                addrs[l] = ((0 * 16 + lg * 4 + r) * ST + cb * 16 + lr) * 2
            pairs.append(write16(addrs))
    acc("P^T strip b16 writes (x2 for dS^T)", pairs)

    # A-frag reads: PTl[(wave*16+lr)*ST + kc2*32+lg*8] b128
    pairs = []
    for kc2 in range(2):
        addrs = {}
        for l in range(64):
            lr, lg = lane_split(l)
            addrs[l] = ((0 * 16 + lr) * ST + kc2 * 32 + lg * 8) * 2
        pairs.append(read128(addrs))
    acc("P^T/dS^T A-frag reads (x2)", pairs)

    # B-frag reads from swizzled transposed images:
    # dOTl[SWZ8(dc*16+lr, kc2*32+lg*8)]
    pairs = []
    for kc2 in range(2):
        for dc in range(DPAD // 16):
            addrs = {}
            for l in range(64):
                lr, lg = lane_split(l)
                addrs[l] = SWZ(dc * 16 + lr, kc2 * 32 + lg * 8, ST) * 2
            pairs.append(read128(addrs))
    acc("dOT/QT swizzled B-frag reads (x2)", pairs)

    print(f"  TOTAL factor (sampled instrs) = {tot[0] / tot[1]:.3f}")

    print("== fa_bwd_dq ==")
    tot = [0, 0]
    # staging writes: Klds/Vlds b128 + KTl scatter — same pattern as dkv
    # dS strip writes: dSw[(lg*4+r)*ST + cb*16+lr] — same as P^T writes
    # A-frag reads dSw[lr*ST + kc2*32+lg*8] — same as above
    # B-frag reads Klds[(cb*16+lr)*SQ + d0] — same as Qlds reads
    # KTl SWZ8 reads — same as dOTl reads
    print("  (patterns identical to dkv rows)")

    print("== fa_fwd ==")
    tot = [0, 0]
    # V^T scatter: Vlds[VSWZ(d0+e, key)] swizzled same as QTl scatter
    # K reads: Klds[(cb*16+lr)*SK + kc*32+lg*8] same as Qlds reads
    # P writes: Pw[(lg*4+r)*SV + cb*16+lr] — SV=72 same as ST; same as P^T writes
    # P reads: Pw[lr*SV + kc2*32+lg*8] same as A-frag reads
    # V^T reads: VSWZ8(dc*16+lr, kc2*32+lg*8) same as dOTl reads
    print("  (patterns identical to dkv rows, SV=SK strides)")


if __name__ == "__main__":
    main()
