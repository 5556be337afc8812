"""GPU parity tests: HIP engine vs CPU oracle, bit-exact.

These are the parity tests proper: every compute result of the engine
(labels, VRF nonce, proofs, verification verdicts) is compared byte-for-byte
with the oracle on the same seeded inputs, at sizes the oracle finishes in
seconds, plus the OpenSSL-pinned golden labels at mainnet N=8192.
All calls cross the C-ABI (include/spacemesh_post.h).
"""
import ctypes
import json
import os

import pytest

import gsm_amd
from oracle import Oracle, Proof, make_meta

pytestmark = pytest.mark.gpu

NODE = bytes([0xA5]) * 32
ATX = bytes([0x5A]) * 32
CHALLENGE = bytes(32)
POW_DIFF = bytes([0x0F]) + bytes([0xFF]) * 31
GOLDEN = json.load(open(os.path.join(os.path.dirname(__file__), "golden",
                                     "golden.json")))


def make_mgr(num_units, lpu, scrypt_n, **kw):
    cfg = gsm_amd.PostConfig(min_num_units=1, labels_per_unit=lpu,
                             k1=12, k2=8, k3=4, pow_difficulty=POW_DIFF)
    kw.setdefault("scratch_bytes", 32 << 30)  # keep test allocations modest
    opts = gsm_amd.PostSetupOpts(num_units=num_units, scrypt_n=scrypt_n,
                                 **kw)
    mgr = gsm_amd.PostSetupManager(NODE, ATX, cfg, opts)
    return cfg, mgr


@pytest.fixture(scope="module")
def small_init():
    """2^14 labels at N=128 in sink mode, labels kept on device."""
    cfg, mgr = make_mgr(1, 1 << 14, 128)
    mgr.prepare_initializer()
    mgr.start_session()
    yield cfg, mgr
    mgr.reset()


def test_labels_bit_exact_vs_oracle(small_init):
    _, mgr = small_init
    total = 1 << 14
    got = mgr.copy_labels(0, total)
    o = Oracle()
    commit = o.commitment(NODE, ATX)
    want, _ = o.init_range(commit, 0, total, 128)
    assert got == want  # byte-for-byte postdata content parity


def test_vrf_nonce_matches_oracle(small_init):
    _, mgr = small_init
    o = Oracle()
    commit = o.commitment(NODE, ATX)
    _, best = o.init_range(commit, 0, 1 << 14, 128)
    got = mgr.vrf_nonce()
    assert got is not None
    assert got[0] == best.index
    assert got[1] == bytes(best.label)


def test_progress_counter_complete(small_init):
    _, mgr = small_init
    st = mgr.status()
    assert st["state"] == mgr.COMPLETE
    assert st["num_labels_written"] == 1 << 14


def test_labels_golden_openssl_n8192():
    """Mainnet scrypt-N: engine labels == OpenSSL-computed golden vectors."""
    lv = GOLDEN["labels_openssl"]
    cfg, mgr = make_mgr(1, 1 << 11, 8192)
    mgr.prepare_initializer()
    mgr.start_session()
    got = mgr.copy_labels(0, 1 << 11)
    for v in lv["labels"]:
        if v["N"] == 8192 and v["index"] < (1 << 11):
            lab = got[v["index"] * 16:(v["index"] + 1) * 16]
            assert lab.hex() == v["full"][:32], v
    mgr.reset()


def test_file_mode_and_resume(tmp_path):
    """postdata_*.bin file split + resume-after-interrupt parity
    (activation/post.go:56,267-271)."""
    d = str(tmp_path)
    lpu, n = 1 << 12, 128
    per_file_labels = 1 << 10
    cfg, mgr = make_mgr(1, lpu, n, data_dir=d,
                        max_file_size=per_file_labels * 16)
    mgr.prepare_initializer()
    mgr.start_session()
    files = sorted(f for f in os.listdir(d) if f.startswith("postdata_")
                   and f.endswith(".bin"))
    assert len(files) == lpu // per_file_labels
    data = b"".join(open(os.path.join(d, f), "rb").read() for f in files)
    o = Oracle()
    commit = o.commitment(NODE, ATX)
    want, _ = o.init_range(commit, 0, lpu, n)
    assert data == want
    assert os.path.exists(os.path.join(d, "postdata_metadata.json"))
    mgr.reset()

    # truncate the last file mid-way; a fresh session must resume and
    # reproduce identical bytes
    last = os.path.join(d, files[-1])
    with open(last, "r+b") as f:
        f.truncate(100 * 16)
    cfg, mgr2 = make_mgr(1, lpu, n, data_dir=d,
                         max_file_size=per_file_labels * 16)
    mgr2.prepare_initializer()
    st = mgr2.status()
    assert st["num_labels_written"] == lpu - per_file_labels + 100
    mgr2.start_session()
    data2 = b"".join(open(os.path.join(d, f), "rb").read() for f in files)
    assert data2 == want
    mgr2.reset()


@pytest.fixture(scope="module")
def roundtrip():
    """init -> prove on GPU; oracle proves the same labels for comparison."""
    NU, LPU, N = 2, 1 << 10, 32
    total = NU * LPU
    cfg, mgr = make_mgr(NU, LPU, N)
    mgr.prepare_initializer()
    mgr.start_session()
    labels = mgr.copy_labels(0, total)
    proof = gsm_amd.api.prove_buffer(
        labels, total, NODE, ATX, CHALLENGE,
        gsm_amd.PostConfig(k1=12, k2=8, k3=4, pow_difficulty=POW_DIFF),
        gsm_amd.ProveOpts(nonces=16))
    yield NU, LPU, N, labels, proof, mgr
    mgr.reset()


def test_proof_bit_exact_vs_oracle(roundtrip):
    NU, LPU, N, labels, proof, _ = roundtrip
    o = Oracle()
    op = o.prove(labels, NU * LPU, CHALLENGE, 12, 8, 16, POW_DIFF)
    assert proof.nonce == op.nonce
    assert proof.pow == op.pow
    assert proof.indices == bytes(op.indices[:op.indices_len])


def test_gpu_verify_accepts_gpu_proof(roundtrip):
    NU, LPU, N, labels, proof, _ = roundtrip
    cfg = gsm_amd.PostConfig(k1=12, k2=8, k3=8, pow_difficulty=POW_DIFF)
    ver = gsm_amd.PostVerifier(cfg, scrypt_n=N)
    meta = gsm_amd.PostProofMetadata(NODE, ATX, CHALLENGE, NU, LPU)
    ver.verify(proof, meta)  # full K2
    # subset + selected-index paths
    cfg3 = gsm_amd.PostConfig(k1=12, k2=8, k3=3, pow_difficulty=POW_DIFF)
    ver3 = gsm_amd.PostVerifier(cfg3, scrypt_n=N)
    ver3.verify(proof, meta,
                gsm_amd.VerifyOpts(subset_seed=b"peer-seed"))
    ver.verify(proof, meta, gsm_amd.VerifyOpts(selected_index=0))


def test_gpu_verify_verdicts_match_oracle(roundtrip):
    """Accept/reject parity incl. the adversarial corrupt-index case
    (systest distributed_post_verification_test.go:253-267)."""
    NU, LPU, N, labels, proof, _ = roundtrip
    o = Oracle()
    meta = gsm_amd.PostProofMetadata(NODE, ATX, CHALLENGE, NU, LPU)
    ometa = make_meta(NODE, ATX, CHALLENGE, NU, LPU)
    cfg = gsm_amd.PostConfig(k1=12, k2=8, k3=8, pow_difficulty=POW_DIFF)
    ver = gsm_amd.PostVerifier(cfg, scrypt_n=N)

    def o_proof(p):
        op = Proof()
        op.nonce = p.nonce
        op.pow = p.pow
        op.num_indices = 8
        op.indices_len = len(p.indices)
        for i, b in enumerate(p.indices):
            op.indices[i] = b
        return op

    # corrupt each byte of the indices in turn; engine and oracle must agree
    agree = 0
    for pos in range(len(proof.indices)):
        bad = bytearray(proof.indices)
        bad[pos] ^= 0x5A
        bp = gsm_amd.PostProof(proof.nonce, bytes(bad), proof.pow)
        orc, _ = o.verify(o_proof(bp), ometa, N, 12, 8, 8, None, -1, POW_DIFF)
        try:
            ver.verify(bp, meta)
            erc = 0
        except gsm_amd.EngineError as e:
            erc = {gsm_amd.api.Status.INVALID_INDEX: 1,
                   gsm_amd.api.Status.POW: 2,
                   gsm_amd.api.Status.INVALID_ARGS: 3}.get(e.code, -1)
        assert (orc == 0) == (erc == 0), (pos, orc, erc)
        if orc == erc:
            agree += 1
    assert agree >= len(proof.indices) - 1  # allow err-code drift, not verdict

    # bad pow
    bp = gsm_amd.PostProof(proof.nonce, proof.indices, proof.pow + 1)
    with pytest.raises(gsm_amd.EngineError) as ei:
        ver.verify(bp, meta)
    assert ei.value.code == gsm_amd.api.Status.POW


def test_verify_batch_mixed(roundtrip):
    NU, LPU, N, labels, proof, _ = roundtrip
    meta = gsm_amd.PostProofMetadata(NODE, ATX, CHALLENGE, NU, LPU)
    bad = gsm_amd.PostProof(proof.nonce,
                            bytes([proof.indices[0] ^ 0x5A]) +
                            proof.indices[1:], proof.pow)
    cfg = gsm_amd.PostConfig(k1=12, k2=8, k3=8, pow_difficulty=POW_DIFF)
    ver = gsm_amd.PostVerifier(cfg, scrypt_n=N)
    res = ver.verify_batch([proof, bad, proof], [meta, meta, meta])
    assert res[0][0] == gsm_amd.api.Status.OK
    assert res[1][0] != gsm_amd.api.Status.OK
    assert res[2][0] == gsm_amd.api.Status.OK


def test_vrf_nonce_verify_gpu(roundtrip):
    NU, LPU, N, labels, proof, mgr = roundtrip
    nonce = mgr.vrf_nonce()
    assert nonce is not None
    cfg = gsm_amd.PostConfig(k1=12, k2=8, k3=8, pow_difficulty=POW_DIFF)
    ver = gsm_amd.PostVerifier(cfg, scrypt_n=N)
    meta = gsm_amd.PostProofMetadata(NODE, ATX, CHALLENGE, NU, LPU)
    ver.verify_vrf_nonce(meta, nonce[0])


def test_cancel_mid_init():
    """Cancellation maps to the Stopped state (post.go:297-301)."""
    import threading
    cfg, mgr = make_mgr(1, 1 << 22, 8192, scratch_bytes=8 << 30)
    mgr.prepare_initializer()
    t = threading.Timer(0.3, mgr.stop)
    t.start()
    with pytest.raises(gsm_amd.EngineError) as ei:
        mgr.start_session()
    assert ei.value.code == gsm_amd.api.Status.CANCELLED
    assert mgr.status()["state"] == mgr.STOPPED
    t.cancel()
    mgr.reset()


def test_providers_and_benchmark():
    eng = gsm_amd.Engine()
    provs = eng.providers()
    assert len(provs) >= 1
    assert provs[0]["memory_bytes"] > 0
    lps = eng.benchmark(0, 8192)
    assert lps > 0
    print("benchmark labels/s:", lps)


def test_prove_from_datadir_roundtrip(tmp_path):
    """File-backed proving: init to postdata_*.bin, prove from the
    directory (the post-service role over a real data dir), verify."""
    d = str(tmp_path)
    NU, LPU, N = 1, 1 << 11, 32
    cfg, mgr = make_mgr(NU, LPU, N, data_dir=d,
                        max_file_size=(1 << 10) * 16)
    mgr.prepare_initializer()
    mgr.start_session()
    mgr.reset()
    pcfg = gsm_amd.PostConfig(min_num_units=1, labels_per_unit=LPU, k1=12,
                              k2=8, k3=8, pow_difficulty=POW_DIFF)
    proof = gsm_amd.api.prove_dir(d, CHALLENGE, pcfg,
                                  gsm_amd.ProveOpts(nonces=16))
    ver = gsm_amd.PostVerifier(pcfg, scrypt_n=N)
    meta = gsm_amd.PostProofMetadata(NODE, ATX, CHALLENGE, NU, LPU)
    ver.verify(proof, meta)
    # and the oracle agrees with the file-backed proof
    labels = b"".join(
        open(os.path.join(d, f), "rb").read()
        for f in sorted(os.listdir(d))
        if f.startswith("postdata_") and f.endswith(".bin"))
    o = Oracle()
    op = o.prove(labels, NU * LPU, CHALLENGE, 12, 8, 16, POW_DIFF)
    assert proof.nonce == op.nonce
    assert proof.indices == bytes(op.indices[:op.indices_len])


def test_high_index_labels_golden():
    """Label indices above 2^32 exercise the high word of LE64(index) in
    the device PBKDF2 password — pinned by the OpenSSL golden vector at
    index 2^32+17 (N=8192)."""
    hi = next(v for v in GOLDEN["labels_openssl"]["labels"]
              if v["N"] == 8192 and v["index"] > 2**32)
    start = 2**32
    cfg, mgr = make_mgr(2, 2**32, 8192, index_start=start,
                        index_end=start + 64)
    mgr.prepare_initializer()
    mgr.start_session()
    got = mgr.copy_labels(0, 64)
    off = hi["index"] - start
    assert got[off * 16:(off + 1) * 16].hex() == hi["full"][:32]
    mgr.reset()


def test_ragged_range_parity():
    """Non-power-of-two, non-wave-aligned batch sizes vs the oracle."""
    total = 1000  # not a multiple of 64 quads or 128 slots
    cfg, mgr = make_mgr(1, total, 128)
    mgr.prepare_initializer()
    mgr.start_session()
    got = mgr.copy_labels(0, total)
    o = Oracle()
    commit = o.commitment(NODE, ATX)
    want, best = o.init_range(commit, 0, total, 128)
    assert got == want
    nonce = mgr.vrf_nonce()
    assert nonce is not None and nonce[0] == best.index
    mgr.reset()


def test_shard_union_equals_whole(tmp_path):
    """Three shard sessions over [0,T) produce byte-identical labels to a
    single whole-range session, and the merged nonce equals the whole-range
    nonce (the 8-GPU sharding axis)."""
    import importlib
    sharding = importlib.import_module("go-spacemesh_amd.sharding")
    T, N = 3000, 128
    whole_cfg, whole = make_mgr(1, T, N)
    whole.prepare_initializer()
    whole.start_session()
    ref_labels = whole.copy_labels(0, T)
    ref_nonce = whole.vrf_nonce()
    whole.reset()

    parts = []
    cands = []
    for r in range(3):
        s, e = sharding.shard_range(T, 3, r)
        cfg, mgr = make_mgr(1, T, N, index_start=s, index_end=e)
        mgr.prepare_initializer()
        mgr.start_session()
        parts.append(mgr.copy_labels(0, e - s))
        n = mgr.vrf_nonce()
        cands.append((n[0], n[1]) if n else None)
        mgr.reset()
    assert b"".join(parts) == ref_labels
    merged = sharding.merge_nonces(cands)
    assert merged is not None and ref_nonce is not None
    assert merged[0] == ref_nonce[0]


def test_random_far_ranges_match_oracle():
    """Size-independent property at BASELINE's full index space: arbitrary
    far shard ranges (up to 2^36) produce exactly the oracle's labels.
    The full 256-GiB space can't be held, but label i depends only on
    (commitment, i), so spot ranges pin the whole space."""
    import random
    rng = random.Random(31)
    o = Oracle()
    commit = o.commitment(NODE, ATX)
    total_units, lpu = 16, 1 << 32  # 2^36 labels, SURVEY cfg2/3 scale
    for _ in range(3):
        start = rng.randrange(0, total_units * lpu - 256)
        cfg, mgr = make_mgr(total_units, lpu, 8192, index_start=start,
                            index_end=start + 256)
        mgr.prepare_initializer()
        mgr.start_session()
        got = mgr.copy_labels(0, 256)
        mgr.reset()
        for off in (0, 97, 255):
            want = o.label(commit, start + off, 8192)[:16]
            assert got[off * 16:(off + 1) * 16] == want, (start, off)


def test_verify_malformed_shapes():
    """Shape checks precede GPU work: wrong indices length and k2 mismatch
    are INVALID_ARGS, not crashes (validation of the wire cap and
    bits-per-index, wire_v1.go:43)."""
    cfg = gsm_amd.PostConfig(min_num_units=1, labels_per_unit=1 << 10,
                             k1=12, k2=8, k3=8, pow_difficulty=POW_DIFF)
    ver = gsm_amd.PostVerifier(cfg, scrypt_n=32)
    meta = gsm_amd.PostProofMetadata(NODE, ATX, CHALLENGE, 1, 1 << 10)
    for bad_indices in [b"", b"\x00" * 3, b"\x00" * 100]:
        with pytest.raises(gsm_amd.EngineError) as ei:
            ver.verify(gsm_amd.PostProof(0, bad_indices, 0), meta)
        assert ei.value.code in (gsm_amd.api.Status.INVALID_ARGS,
                                 gsm_amd.api.Status.POW)


def test_post_service_real_prover_roundtrip(tmp_path):
    """Out-of-process placement: node server + real post-service child
    proving over gRPC from a GPU-initialized data dir (the post-service
    drop-in role)."""
    import importlib
    sup_mod = importlib.import_module("go-spacemesh_amd.supervisor")
    d = str(tmp_path)
    NU, LPU, N = 1, 1 << 12, 32
    cfg, mgr = make_mgr(NU, LPU, N, data_dir=d, max_file_size=1 << 16)
    mgr.prepare_initializer()
    mgr.start_session()
    mgr.reset()

    server = sup_mod.PostServiceServer()
    sup = sup_mod.PostSupervisor(server.address, d, nonces=16, k1=26, k2=37)
    sup.start()
    try:
        client = server.wait_for_client(timeout=30, poll_interval=0.2)
        info = client.info()
        assert info.node_id == NODE and info.num_units == NU
        proof = client.proof(CHALLENGE, timeout=120)
        # verify through the engine like the node's validator would
        pcfg = gsm_amd.PostConfig(min_num_units=1, labels_per_unit=LPU,
                                  k1=26, k2=37, k3=37)
        ver = gsm_amd.PostVerifier(pcfg, scrypt_n=N)
        meta = gsm_amd.PostProofMetadata(NODE, ATX, CHALLENGE, NU, LPU)
        ver.verify(gsm_amd.PostProof(proof.nonce, proof.indices, proof.pow),
                   meta)
    finally:
        sup.stop()
        server.stop()


def test_concurrent_verify_threads(roundtrip):
    """The SAFETY contract (post_verifier.go:227): verify must be callable
    concurrently from N workers."""
    from concurrent.futures import ThreadPoolExecutor
    NU, LPU, N, labels, proof, _ = roundtrip
    cfg = gsm_amd.PostConfig(k1=12, k2=8, k3=8, pow_difficulty=POW_DIFF)
    ver = gsm_amd.PostVerifier(cfg, scrypt_n=N)
    meta = gsm_amd.PostProofMetadata(NODE, ATX, CHALLENGE, NU, LPU)

    def one(i):
        if i % 3 == 2:  # mix in invalid proofs
            bad = gsm_amd.PostProof(proof.nonce, proof.indices, proof.pow + 1)
            try:
                ver.verify(bad, meta)
                return "accepted-bad"
            except gsm_amd.EngineError as e:
                return "rejected" if e.code == gsm_amd.api.Status.POW \
                    else f"wrong-{e.code.name}"
        ver.verify(proof, meta)
        return "ok"

    with ThreadPoolExecutor(8) as ex:
        results = list(ex.map(one, range(24)))
    assert results.count("ok") == 16
    assert results.count("rejected") == 8


def test_concurrent_mixed_ops(roundtrip, tmp_path):
    """Whole-ABI concurrency contract (the strongest form available with
    no Go toolchain to compile the cgo shim): one thread
    runs an init session, one proves, N verify, one polls
    providers/benchmark/selftests — all against the same engine library at
    once, like a node initializing one identity while gossip-verifying and
    self-proving another."""
    from concurrent.futures import ThreadPoolExecutor
    NU, LPU, N, labels, proof, _ = roundtrip
    vcfg = gsm_amd.PostConfig(k1=12, k2=8, k3=8, pow_difficulty=POW_DIFF)
    meta = gsm_amd.PostProofMetadata(NODE, ATX, CHALLENGE, NU, LPU)

    def do_init():
        cfg, mgr = make_mgr(1, 1 << 12, 128, data_dir=str(tmp_path),
                            max_file_size=1 << 15)
        mgr.prepare_initializer()
        mgr.start_session()
        st = mgr.status()
        mgr.reset()
        return ("init", st["num_labels_written"] == 1 << 12)

    def do_prove():
        pcfg = gsm_amd.PostConfig(min_num_units=1, labels_per_unit=LPU,
                                  k1=12, k2=8, k3=4,
                                  pow_difficulty=POW_DIFF)
        pr = gsm_amd.api.prove_buffer(labels, NU * LPU, NODE, ATX,
                                      CHALLENGE, pcfg,
                                      gsm_amd.ProveOpts(nonces=16))
        return ("prove", pr.nonce == proof.nonce and
                pr.indices == proof.indices)

    def do_verify(i):
        ver = gsm_amd.PostVerifier(vcfg, scrypt_n=N)
        ver.verify(proof, meta)
        return ("verify", True)

    def do_poll():
        eng = gsm_amd.api.Engine()
        provs = eng.providers()
        b = eng.benchmark(provider_id=0, scrypt_n=128)
        h = eng.selftest_blake3(b"concurrency")
        lab = eng.selftest_label(NODE, ATX, 5, 128)
        o = Oracle()
        commit = o.commitment(NODE, ATX)
        want, _ = o.init_range(commit, 5, 6, 128)
        return ("poll", len(provs) >= 1 and b > 0 and len(h) == 32 and
                lab[:16] == want)

    jobs = [do_init, do_prove, do_poll] + [lambda i=i: do_verify(i)
                                           for i in range(5)]
    with ThreadPoolExecutor(8) as ex:
        futs = [ex.submit(j) for j in jobs]
        results = [f.result(timeout=300) for f in futs]
    assert all(ok for _, ok in results), results


def test_scan_hit_overflow_is_loud():
    """A pathological K1/num_labels configuration that makes every label
    pass every nonce must fail loudly (scan hit-buffer overflow), never
    return a silently-truncated proof."""
    total = 1 << 21
    cfg, mgr = make_mgr(1, total, 32)
    mgr.prepare_initializer()
    mgr.start_session()
    labels = mgr.copy_labels(0, total)
    mgr.reset()
    # k1 == num_labels -> proving difficulty ~2^64: all 288*2 nonce values
    # of every label hit = 1.2e9 hits >> the 4M hit buffer
    bad = gsm_amd.PostConfig(min_num_units=1, labels_per_unit=total,
                             k1=total, k2=37, pow_difficulty=POW_DIFF)
    with pytest.raises(gsm_amd.EngineError) as ei:
        gsm_amd.api.prove_buffer(labels, total, NODE, ATX, CHALLENGE, bad,
                                 gsm_amd.ProveOpts(nonces=288))
    assert "overflow" in str(ei.value)


def test_selfcheck_period_preserves_parity():
    """POST_SELFCHECK_PERIOD thins the host reference-label check (multi-
    rank contention knob) but must not change any output byte."""
    os.environ["POST_SELFCHECK_PERIOD"] = "4"
    try:
        cfg, mgr = make_mgr(1, 1 << 13, 128)
        mgr.prepare_initializer()
        mgr.start_session()
        got = mgr.copy_labels(0, 1 << 13)
        mgr.reset()
    finally:
        del os.environ["POST_SELFCHECK_PERIOD"]
    o = Oracle()
    commit = o.commitment(NODE, ATX)
    want, _ = o.init_range(commit, 0, 1 << 13, 128)
    assert got == want


def test_scan_kernel_variants_agree(roundtrip):
    """All three scan kernels (shared T-table, bank-replicated Te0,
    4-table interleaved) must produce the identical proof on the same
    labels — they implement one algorithm with different LDS layouts."""
    NU, LPU, N, labels, proof, _ = roundtrip
    cfg = gsm_amd.PostConfig(k1=12, k2=8, k3=4, pow_difficulty=POW_DIFF)
    results = {}
    for mode in ["shared", "bankrep", "tt4"]:
        os.environ["POST_SCAN_MODE"] = mode
        try:
            pr = gsm_amd.api.prove_buffer(labels, NU * LPU, NODE, ATX,
                                          CHALLENGE, cfg,
                                          gsm_amd.ProveOpts(nonces=16))
            results[mode] = (pr.nonce, pr.indices, pr.pow)
        finally:
            del os.environ["POST_SCAN_MODE"]
    assert results["shared"] == (proof.nonce, proof.indices, proof.pow)
    assert results["bankrep"] == results["shared"]
    assert results["tt4"] == results["shared"]


def test_cfg1_checksum_regression():
    """Determinism pin across kernel changes: sha256 over the first 2^20
    labels of BASELINE config 1 (mainnet N, fixed identity).  The absolute
    values are pinned by the OpenSSL golden spot labels; this checksums the
    whole prefix.  Constant generated on MI355X (r01)."""
    import hashlib
    cfg, mgr = make_mgr(1, 1 << 20, 8192)
    mgr.prepare_initializer()
    mgr.start_session()
    got = mgr.copy_labels(0, 1 << 20)
    mgr.reset()
    digest = hashlib.sha256(got).hexdigest()
    # recorded on MI355X, round 1 (gpurun summary11)
    expected = ("3cb88b0552f172abbf82a210509c3632"
                "a1155df747b825b82334b67173219c4b")
    assert digest == expected


def test_session_cycle_no_leak():
    """Create/run/free sessions and verify batches repeatedly; device free
    memory must return to (near) its starting level — no leaked scratch."""
    import torch
    free0, _ = torch.cuda.mem_get_info(0)
    for i in range(5):
        cfg, mgr = make_mgr(1, 1 << 12, 128)
        mgr.prepare_initializer()
        mgr.start_session()
        labels = mgr.copy_labels(0, 1 << 12)
        proof = gsm_amd.api.prove_buffer(
            labels, 1 << 12, NODE, ATX, CHALLENGE,
            gsm_amd.PostConfig(min_num_units=1, labels_per_unit=1 << 12,
                               k1=12, k2=8, pow_difficulty=POW_DIFF),
            gsm_amd.ProveOpts(nonces=16))
        mgr.reset()
        ver = gsm_amd.PostVerifier(
            gsm_amd.PostConfig(min_num_units=1, labels_per_unit=1 << 12,
                               k1=12, k2=8, k3=8, pow_difficulty=POW_DIFF),
            scrypt_n=128)
        meta = gsm_amd.PostProofMetadata(NODE, ATX, CHALLENGE, 1, 1 << 12)
        ver.verify(proof, meta)
    free1, _ = torch.cuda.mem_get_info(0)
    # the cached verify workspace and AES tables stay resident (bounded);
    # everything else must be returned
    assert free0 - free1 < (2 << 30), (free0, free1)


def test_sharded_prove_equals_whole(roundtrip):
    """Two index-range scan shards merged on the host produce the exact
    proof of the single-GPU prover (the 8-GPU proving axis)."""
    import importlib
    proving = importlib.import_module("go-spacemesh_amd.proving")
    NU, LPU, N, labels, proof, _ = roundtrip
    total = NU * LPU
    cfg = gsm_amd.PostConfig(min_num_units=1, labels_per_unit=LPU,
                             k1=12, k2=8, pow_difficulty=POW_DIFF)
    cut = 700 * 16  # ragged split
    merged = proving.prove_sharded(
        [(labels[:cut], 0), (labels[cut:], 700)], total, CHALLENGE, cfg,
        nonces=16)
    assert merged is not None
    assert merged.nonce == proof.nonce
    assert merged.pow == proof.pow
    assert merged.indices == proof.indices


def test_romix2_env_path_parity(tmp_path):
    """The POST_ROMIX2=1 dual-stream kernel path stays bit-exact (guards
    the alternate path against rot).  Runs in a subprocess because the
    launcher caches the env choice per process."""
    import subprocess
    import sys
    script = r"""
import sys
sys.path.insert(0, %r)
sys.path.insert(0, %r)
import gsm_amd
from oracle import Oracle
NODE, ATX = bytes([0xA5])*32, bytes([0x5A])*32
POW = bytes([0x0F]) + bytes([0xFF])*31
cfg = gsm_amd.PostConfig(min_num_units=1, labels_per_unit=1500, k1=12,
                         k2=8, k3=4, pow_difficulty=POW)
opts = gsm_amd.PostSetupOpts(num_units=1, scrypt_n=128,
                             scratch_bytes=8 << 30)
mgr = gsm_amd.PostSetupManager(NODE, ATX, cfg, opts)
mgr.prepare_initializer()
mgr.start_session()
got = mgr.copy_labels(0, 1500)
o = Oracle()
want, best = o.init_range(o.commitment(NODE, ATX), 0, 1500, 128)
assert got == want, "labels mismatch under POST_ROMIX2"
n = mgr.vrf_nonce()
assert n is not None and n[0] == best.index
mgr.reset()
print("romix2 parity OK")
"""
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ, POST_ROMIX2="1")
    r = subprocess.run(
        [sys.executable, "-c", script % (repo, os.path.join(repo, "oracle"))],
        capture_output=True, text=True, timeout=240, env=env)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "romix2 parity OK" in r.stdout


def test_randomized_config_fuzz_parity():
    """Randomized-config round-trip fuzz (the reference runs fuzzing over
    its codecs/state, Makefile:195; here: random scrypt-N / space sizes /
    K-params, engine vs oracle end to end, fixed seed)."""
    import random
    rng = random.Random(0xF00D)
    o = Oracle()
    for trial in range(5):
        n = 1 << rng.randrange(1, 10)         # scrypt N in 2..512
        total = rng.randrange(200, 2500)
        k2 = rng.randrange(4, 16)
        k1 = k2 + rng.randrange(0, 8)
        nonces = 16 * rng.randrange(1, 3)
        node = bytes(rng.randrange(256) for _ in range(32))
        atx = bytes(rng.randrange(256) for _ in range(32))
        challenge = bytes(rng.randrange(256) for _ in range(32))
        cfg = gsm_amd.PostConfig(min_num_units=1, labels_per_unit=total,
                                 k1=k1, k2=k2, k3=k2,
                                 pow_difficulty=POW_DIFF)
        opts = gsm_amd.PostSetupOpts(num_units=1, scrypt_n=n,
                                     scratch_bytes=8 << 30)
        mgr = gsm_amd.PostSetupManager(node, atx, cfg, opts)
        mgr.prepare_initializer()
        mgr.start_session()
        got = mgr.copy_labels(0, total)
        nonce = mgr.vrf_nonce()
        mgr.reset()
        commit = o.commitment(node, atx)
        want, best = o.init_range(commit, 0, total, n)
        assert got == want, (trial, n, total)
        assert nonce is not None and nonce[0] == best.index, (trial,)
        try:
            proof = gsm_amd.api.prove_buffer(
                got, total, node, atx, challenge, cfg,
                gsm_amd.ProveOpts(nonces=nonces))
        except gsm_amd.EngineError as e:
            assert e.code == gsm_amd.api.Status.NO_NONCE
            with pytest.raises(ValueError):
                o.prove(got, total, challenge, k1, k2, nonces, POW_DIFF)
            continue
        op = o.prove(got, total, challenge, k1, k2, nonces, POW_DIFF)
        assert (proof.nonce, proof.pow) == (op.nonce, op.pow), (trial,)
        assert proof.indices == bytes(op.indices[:op.indices_len])
        ver = gsm_amd.PostVerifier(cfg, scrypt_n=n)
        meta = gsm_amd.PostProofMetadata(node, atx, challenge, 1, total)
        ver.verify(proof, meta)


def test_engine_reproduces_frozen_proof_fixture():
    """The engine end-to-end reproduces the committed protocol fixture
    (tests/golden 'protocol_frozen'): init checksum, VRF nonce, proof."""
    import hashlib
    fx = GOLDEN["protocol_frozen"]["proof_fixture"]
    node = bytes.fromhex(fx["node_id"])
    atx = bytes.fromhex(fx["atx_id"])
    cfg = gsm_amd.PostConfig(
        min_num_units=1, labels_per_unit=fx["num_labels"], k1=fx["k1"],
        k2=fx["k2"], k3=fx["k2"],
        pow_difficulty=bytes.fromhex(fx["pow_difficulty"]))
    opts = gsm_amd.PostSetupOpts(num_units=1, scrypt_n=fx["scrypt_n"],
                                 scratch_bytes=4 << 30)
    mgr = gsm_amd.PostSetupManager(node, atx, cfg, opts)
    mgr.prepare_initializer()
    mgr.start_session()
    labels = mgr.copy_labels(0, fx["num_labels"])
    assert hashlib.sha256(labels).hexdigest() == fx["labels_sha256"]
    nonce = mgr.vrf_nonce()
    assert nonce is not None and nonce[0] == fx["vrf_nonce_index"]
    mgr.reset()
    proof = gsm_amd.api.prove_buffer(
        labels, fx["num_labels"], node, atx, bytes.fromhex(fx["challenge"]),
        cfg, gsm_amd.ProveOpts(nonces=fx["nonces"]))
    assert proof.nonce == fx["proof_nonce"]
    assert proof.pow == fx["proof_pow"]
    assert proof.indices.hex() == fx["proof_indices"]


def test_verify_batch_commitments_survive_early_rejects(roundtrip):
    """Regression: a proof rejected at the pow/shape stage must not skew the
    commitment table of the surviving proofs (absolute-index table)."""
    NU, LPU, N, labels, proof, _ = roundtrip
    meta = gsm_amd.PostProofMetadata(NODE, ATX, CHALLENGE, NU, LPU)
    bad_pow = gsm_amd.PostProof(proof.nonce, proof.indices, proof.pow + 1)
    bad_shape = gsm_amd.PostProof(proof.nonce, proof.indices[:-2], proof.pow)
    cfg = gsm_amd.PostConfig(k1=12, k2=8, k3=8, pow_difficulty=POW_DIFF)
    ver = gsm_amd.PostVerifier(cfg, scrypt_n=N)
    res = ver.verify_batch([bad_pow, proof, bad_shape, proof],
                           [meta, meta, meta, meta])
    assert res[0][0] == gsm_amd.api.Status.POW
    assert res[1][0] == gsm_amd.api.Status.OK
    assert res[2][0] == gsm_amd.api.Status.INVALID_ARGS
    assert res[3][0] == gsm_amd.api.Status.OK


def test_per_proof_seeds_match_oracle(roundtrip):
    """Seeded batch verification: each proof sampled with its own seed must
    agree with the oracle's per-seed verdicts (the gossip shape —
    per-peer Subset seeds, validation.go:206-209)."""
    NU, LPU, N, labels, proof, _ = roundtrip
    o = Oracle()
    ometa = make_meta(NODE, ATX, CHALLENGE, NU, LPU)
    cfg = gsm_amd.PostConfig(k1=12, k2=8, k3=3, pow_difficulty=POW_DIFF)
    ver = gsm_amd.PostVerifier(cfg, scrypt_n=N)
    meta = gsm_amd.PostProofMetadata(NODE, ATX, CHALLENGE, NU, LPU)
    seeds = [bytes([i]) * 32 for i in range(8)]
    res = ver.verify_batch([proof] * 8, [meta] * 8, seeds=seeds)
    assert all(s == gsm_amd.api.Status.OK for s, _ in res)

    def o_proof(p):
        op = Proof()
        op.nonce = p.nonce
        op.pow = p.pow
        op.num_indices = 8
        op.indices_len = len(p.indices)
        for i, b in enumerate(p.indices):
            op.indices[i] = b
        return op

    # corrupt one byte; per-seed verdicts must match the oracle exactly
    bad = bytearray(proof.indices)
    bad[2] ^= 0x3C
    bp = gsm_amd.PostProof(proof.nonce, bytes(bad), proof.pow)
    res = ver.verify_batch([bp] * 8, [meta] * 8, seeds=seeds)
    for seed, (status, inv) in zip(seeds, res):
        orc, oinv = o.verify(o_proof(bp), ometa, N, 12, 8, 3, seed, -1,
                             POW_DIFF)
        assert (orc == 0) == (status == gsm_amd.api.Status.OK), seed
        if orc == 1:
            assert inv == oinv


def test_batching_verifier_end_to_end_gpu(roundtrip):
    """BatchingVerifier over the real engine: concurrent per-seed verifies
    grouped into engine batches."""
    NU, LPU, N, labels, proof, _ = roundtrip
    from concurrent.futures import ThreadPoolExecutor
    cfg = gsm_amd.PostConfig(k1=12, k2=8, k3=3, pow_difficulty=POW_DIFF)
    ver = gsm_amd.PostVerifier(cfg, scrypt_n=N)
    meta = gsm_amd.PostProofMetadata(NODE, ATX, CHALLENGE, NU, LPU)
    bv = gsm_amd.BatchingVerifier(ver, max_batch=32, max_wait_s=0.01)
    with ThreadPoolExecutor(8) as ex:
        list(ex.map(lambda i: bv.verify(proof, meta, bytes([i]) * 32),
                    range(24)))
    bv.close()


def test_datadir_identity_guard(tmp_path):
    """A postdata dir is bound to one identity/config: re-initializing with
    a different node id / ATX / params must be refused
    (initialization metadata guard; commitment read-back at
    activation/post.go:373-435)."""
    d = str(tmp_path)
    cfg, mgr = make_mgr(1, 1 << 10, 32, data_dir=d, max_file_size=1 << 14)
    mgr.prepare_initializer()
    mgr.start_session()
    mgr.reset()
    # same config resumes fine
    cfg, mgr2 = make_mgr(1, 1 << 10, 32, data_dir=d, max_file_size=1 << 14)
    mgr2.prepare_initializer()
    mgr2.reset()
    # different identity refused
    other = gsm_amd.PostSetupManager(
        bytes([1]) * 32, ATX,
        gsm_amd.PostConfig(min_num_units=1, labels_per_unit=1 << 10, k1=12,
                           k2=8, pow_difficulty=POW_DIFF),
        gsm_amd.PostSetupOpts(num_units=1, scrypt_n=32, data_dir=d,
                              max_file_size=1 << 14,
                              scratch_bytes=8 << 30))
    with pytest.raises(gsm_amd.EngineError) as ei:
        other.prepare_initializer()
    assert ei.value.code == gsm_amd.api.Status.INVALID_ARGS
    # different scrypt N refused
    cfgn, mgrn = make_mgr(1, 1 << 10, 64, data_dir=d,
                          max_file_size=1 << 14)
    with pytest.raises(gsm_amd.EngineError):
        mgrn.prepare_initializer()


def test_nonce_survives_kill_and_resume(tmp_path):
    """The persisted VRF nonce must survive reopening the data dir: a
    sharded session's shard-local minimum feeds the cross-GPU min-reduce
    and cannot be recomputed from the remaining range alone (reference
    initializer keeps metadata Nonce across resume)."""
    d = str(tmp_path)
    lpu = 1 << 12
    # shard [0, lpu/2): complete it, so the shard's minimum is final
    cfg, mgr = make_mgr(1, lpu, 128, data_dir=d, max_file_size=1 << 15,
                        index_start=0, index_end=lpu // 2)
    mgr.prepare_initializer()
    mgr.start_session()
    got = mgr.vrf_nonce()
    assert got is not None
    mgr.reset()
    md = json.load(open(os.path.join(d, "postdata_metadata.json")))
    assert md["Nonce"] == got[0]

    # reopen the same shard: prepare_initializer alone (before any new
    # labeling) must already carry the persisted nonce, and metadata must
    # not lose it to the session-creation rewrite
    cfg, mgr2 = make_mgr(1, lpu, 128, data_dir=d, max_file_size=1 << 15,
                         index_start=0, index_end=lpu // 2)
    mgr2.prepare_initializer()
    md2 = json.load(open(os.path.join(d, "postdata_metadata.json")))
    assert md2.get("Nonce") == got[0]
    assert md2.get("NonceValue") == md["NonceValue"]
    got2 = mgr2.vrf_nonce()
    assert got2 == got
    mgr2.reset()

    # oracle cross-check: the persisted minimum is the true shard minimum
    o = Oracle()
    commit = o.commitment(NODE, ATX)
    _, best = o.init_range(commit, 0, lpu // 2, 128)
    assert got[0] == best.index
    assert got[1] == bytes(best.label)
