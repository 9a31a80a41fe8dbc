"""XCD-aware CU-mask computation for fractional compute on gfx950.

MI355X has 256 CUs in 8 XCDs of 32; each XCD owns a private 4 MiB L2 and its
own share of HBM bandwidth. A naive "first N CUs" mask would pin a 25% pod to
two whole XCDs and leave the other six idle for that pod — skewing both L2
capacity and DVFS behaviour. Instead, masks are dealt **round-robin across
XCDs** so an X% pod gets ≈X% of every XCD's CUs (and therefore ≈X% of
aggregate L2 and memory bandwidth).

The mask format matches ``hsa_amd_queue_cu_set_mask`` (hsa_ext_amd.h:1359):
little-endian array of uint32, bit i = CU i enabled. ROCr requires masks at
**CU-pair granularity** (bits 2k and 2k+1 set together — the header's
"0x33 valid, 0x5/0x6 invalid" rule), so allocation works in pairs: 128 pairs
per GPU, 16 per XCD.
"""
from __future__ import annotations

from typing import List, Tuple

from .. import consts


def cu_count_for_percent(percent: int, total_cus: int = consts.GFX950_CU_COUNT) -> int:
    """gpu-core units (percent of a card) → CU count, minimum one CU pair,
    rounded UP to even (ROCr masks have CU-pair granularity)."""
    if percent >= consts.GPU_PERCENT_EACH_CARD:
        return total_cus
    n = max(1, (percent * total_cus + consts.GPU_PERCENT_EACH_CARD // 2)
            // consts.GPU_PERCENT_EACH_CARD)
    return min(total_cus, n + (n & 1))


def xcd_round_robin_cus(
    n_cus: int,
    total_cus: int = consts.GFX950_CU_COUNT,
    xcd_count: int = consts.GFX950_XCD_COUNT,
    offset: int = 0,
) -> List[int]:
    """Pick ``n_cus`` (rounded up to even) physical CU ids spread evenly
    across XCDs, in whole CU pairs.

    CU ids are laid out XCD-major (CU i lives on XCD i // 32). ``offset``
    (in CUs, rounded to pairs) rotates the starting pair within every XCD so
    that two co-scheduled pods with disjoint offsets get disjoint masks.
    """
    per_xcd = total_cus // xcd_count
    pairs_per_xcd = per_xcd // 2
    n_pairs = (n_cus + 1) // 2
    pair_offset = offset // 2
    base, extra = divmod(n_pairs, xcd_count)
    cus: List[int] = []
    for xcd in range(xcd_count):
        take = base + (1 if xcd < extra else 0)
        for k in range(take):
            pair = (pair_offset + k) % pairs_per_xcd
            cu0 = xcd * per_xcd + pair * 2
            cus.extend((cu0, cu0 + 1))
    return cus


def mask_words_from_cus(cus: List[int], total_cus: int = consts.GFX950_CU_COUNT) -> List[int]:
    words = [0] * ((total_cus + 31) // 32)
    for cu in cus:
        words[cu // 32] |= 1 << (cu % 32)
    return words


def mask_for_percent(
    percent: int,
    total_cus: int = consts.GFX950_CU_COUNT,
    xcd_count: int = consts.GFX950_XCD_COUNT,
    offset: int = 0,
) -> Tuple[List[int], int]:
    """Returns (mask words, cu count)."""
    n = cu_count_for_percent(percent, total_cus)
    cus = xcd_round_robin_cus(n, total_cus, xcd_count, offset)
    return mask_words_from_cus(cus, total_cus), n


def mask_hex(words: List[int]) -> str:
    """Env-var encoding consumed by the HSA interposer (EGPU_CU_MASK):
    comma-separated little-endian hex words."""
    return ",".join(f"{w:08x}" for w in words)


def parse_mask_hex(s: str) -> List[int]:
    return [int(w, 16) for w in s.split(",") if w]


def popcount(words: List[int]) -> int:
    return sum(bin(w).count("1") for w in words)
