"""CU-mask math: XCD-round-robin distribution, percent→CU mapping, hex codec."""
from elastic_gpu_agent_amd.isolation.cumask import (
    cu_count_for_percent,
    mask_for_percent,
    mask_hex,
    mask_words_from_cus,
    parse_mask_hex,
    popcount,
    xcd_round_robin_cus,
)


def test_percent_to_cu_count():
    assert cu_count_for_percent(100) == 256
    assert cu_count_for_percent(150) == 256  # clamped to a full card
    assert cu_count_for_percent(50) == 128
    assert cu_count_for_percent(25) == 64
    assert cu_count_for_percent(1) == 4  # 2.56 → 3 → rounded up to a CU pair
    assert cu_count_for_percent(0) == 2  # never less than one CU pair


def test_round_robin_spreads_across_xcds():
    cus = xcd_round_robin_cus(64)  # 25% of the card
    assert len(cus) == 64
    per_xcd = [0] * 8
    for cu in cus:
        per_xcd[cu // 32] += 1
    assert per_xcd == [8] * 8  # exactly even across all 8 XCDs


def test_round_robin_uneven_spread():
    cus = xcd_round_robin_cus(13)  # 13 CUs → 7 pairs
    per_xcd = [0] * 8
    for cu in cus:
        per_xcd[cu // 32] += 1
    # one pair on each of the first seven XCDs
    assert per_xcd == [2, 2, 2, 2, 2, 2, 2, 0]


def test_pair_granularity():
    cus = xcd_round_robin_cus(64)
    s = set(cus)
    for cu in s:
        assert cu ^ 1 in s  # every CU's pair sibling is present


def test_offset_rotates_within_xcd():
    a = set(xcd_round_robin_cus(64, offset=0))
    b = set(xcd_round_robin_cus(64, offset=8))
    assert not a & b  # disjoint 25% masks at distinct offsets


def test_mask_words_and_hex_roundtrip():
    cus = xcd_round_robin_cus(96)
    words = mask_words_from_cus(cus)
    assert len(words) == 8  # 256 CUs / 32 bits
    assert popcount(words) == 96
    s = mask_hex(words)
    assert parse_mask_hex(s) == words


def test_mask_for_percent_full_card():
    words, n = mask_for_percent(100)
    assert n == 256
    assert popcount(words) == 256
    assert all(w == 0xFFFFFFFF for w in words)
