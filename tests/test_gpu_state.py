"""GPU parity: full synthetic Deneb BeaconState hash_tree_root (C3 proper)
vs the hashlib restatement, at a size the reference runs in seconds."""
import pytest

import ssz_ref

pytestmark = pytest.mark.gpu


def test_beacon_state_root_parity():
    from lighthouse_amd import beacon_state as bs

    st = bs.generate(4096, seed=123)
    got = bs.state_root(st)
    want = ssz_ref.beacon_state_root_ref(st)
    assert got == want


def test_beacon_state_root_field_sensitivity():
    from lighthouse_amd import beacon_state as bs

    st = bs.generate(1024, seed=7)
    r1 = bs.state_root(st)
    st["eth1_deposit_index"] += 1
    r2 = bs.state_root(st)
    assert r1 != r2
