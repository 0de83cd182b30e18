"""Generate the committed golden fixtures under tests/golden/.

Run from the repo root:  python tests/golden/make_golden.py

These fixtures pin the deterministic input-generator spec (dj_rng.h), the
row-hash spec (dj_hash.h) and the oracle's join semantics so that any
accidental drift — in the C oracle, the numpy restatement, or the HIP
kernels — is caught by tests/test_oracle.py and tests/test_gpu_parity.py.

The join-semantics fixtures derive from the reference's own analytical
known-answer tests (multiples-of-3/5 join,
/root/reference/test/compare_against_analytical.cu:44-54), which are
hash-function-independent; the generator/hash fixtures pin OUR documented
spec (the reference's generator is not device-reproducible and its
MurmurHash3 placement is pinned by no reference test — SURVEY.md §8c).
"""
import os
import sys

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", ".."))
import oracle  # noqa: E402

OUT = os.path.dirname(os.path.abspath(__file__))


def main():
    # 1. Generator fixture: first 1024 rows of the global 1M-row build/probe
    #    tables at the benchmark's defaults (rand_max = 2N, selectivity 0.3,
    #    seed 1234 — reference benchmark defaults, benchmark/distributed_join.cu:96-109,187-188).
    n = 1_000_000
    bk, bp = oracle.gen_build(n, nrows=1024)
    pk, pp = oracle.gen_probe(n, n, selectivity=0.3, nrows=1024)
    np.save(os.path.join(OUT, "gen_build_keys_1M_head.npy"), bk)
    np.save(os.path.join(OUT, "gen_probe_keys_1M_head.npy"), pk)

    # 2. Hash fixture: murmur3 (seed 0 / 12345678) and identity of 256 keys.
    keys = np.concatenate([np.arange(128, dtype=np.int64),
                           np.array([2**40 + 7, -1, -2**40, 2**62, -2**62], dtype=np.int64),
                           bk[:123]])
    h0 = np.array([oracle.row_hash(k, oracle.HASH_MURMUR3, 0) for k in keys], dtype=np.uint32)
    h1 = np.array([oracle.row_hash(k, oracle.HASH_MURMUR3, 12345678) for k in keys], dtype=np.uint32)
    hid = np.array([oracle.row_hash(k, oracle.HASH_IDENTITY, 0) for k in keys], dtype=np.uint32)
    np.save(os.path.join(OUT, "hash_keys.npy"), keys)
    np.save(os.path.join(OUT, "hash_murmur3_seed0.npy"), h0)
    np.save(os.path.join(OUT, "hash_murmur3_seed12345678.npy"), h1)
    np.save(os.path.join(OUT, "hash_identity.npy"), hid)

    # 3. Join fixture: 10k x 10k generated join, sorted rows (int64).
    n2 = 10_000
    bk2, bp2 = oracle.gen_build(n2)
    pk2, pp2 = oracle.gen_probe(n2, n2, selectivity=0.3)
    rows = oracle.sort_rows(*oracle.inner_join(bk2, bp2, pk2, pp2))
    np.save(os.path.join(OUT, "join_10k_sorted.npy"), np.stack(rows))

    print("golden fixtures written:", sorted(f for f in os.listdir(OUT) if f.endswith(".npy")))


if __name__ == "__main__":
    main()
