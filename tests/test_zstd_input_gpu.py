"""zstd-compressed INPUT blocks (SURVEY §8f-1 input side): decoded
host-side at load (util/compression.h ZSTD framing), the rest of the
pipeline unchanged; outputs bit-exact vs the oracle.  Output compression
stays none/snappy (GPU zstd encode out of scope, documented)."""
import pytest

import oracle
import toplingdb_amd as dcw

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module", autouse=True)
def gpu():
    dcw.init(0)
    yield
    dcw.shutdown()


def _zstd_run(tmp_path, name, n, seed):
    import random
    rnd = random.Random(seed)
    kvs = sorted({b"k%015d" % rnd.randrange(4 * n) for _ in range(n)})
    es = [(oracle.make_ikey(k, 100 + i, 1), (b"val-%d-" % i) + k)
          for i, k in enumerate(kvs)]
    p = str(tmp_path / name)
    with open(p, "wb") as f:
        f.write(oracle.build_sst(es, oracle.default_table_opts(compression=7)))
    return p


def test_zstd_inputs_bit_exact(tmp_path):
    runs = [[_zstd_run(tmp_path, "z%d.sst" % r, 30000, 5 + r)]
            for r in range(3)]
    og = tmp_path / "g"
    oo = tmp_path / "o"
    og.mkdir()
    oo.mkdir()
    rg = dcw.execute(dcw.make_job(runs, str(og), compression=1,
                                  bottommost_level=1))
    ro = oracle.execute(oracle.make_job(runs, str(oo), compression=1,
                                        bottommost_level=1))
    assert rg["out_entries"] == ro["out_entries"] > 0
    for fg, fo in zip(rg["files"], ro["files"]):
        assert open(fg["path"], "rb").read() == open(fo["path"], "rb").read()


def test_mixed_zstd_snappy_inputs(tmp_path):
    rz = [_zstd_run(tmp_path, "mz.sst", 20000, 11)]
    ps = str(tmp_path / "ms.sst")
    dcw.gen_sst(ps, seed=12, num_entries=20000, seq_base=200000, compression=1)
    og = tmp_path / "g"
    oo = tmp_path / "o"
    og.mkdir()
    oo.mkdir()
    rg = dcw.execute(dcw.make_job([rz, [ps]], str(og), compression=1,
                                  bottommost_level=1))
    ro = oracle.execute(oracle.make_job([rz, [ps]], str(oo), compression=1,
                                        bottommost_level=1))
    for fg, fo in zip(rg["files"], ro["files"]):
        assert open(fg["path"], "rb").read() == open(fo["path"], "rb").read()


def test_corrupt_zstd_input_fails(tmp_path):
    p = _zstd_run(tmp_path, "c.sst", 20000, 13)
    blob = bytearray(open(p, "rb").read())
    blob[len(blob) // 3] ^= 0xFF
    open(p, "wb").write(bytes(blob))
    out = tmp_path / "out"
    out.mkdir()
    with pytest.raises(RuntimeError):
        dcw.execute(dcw.make_job([[p]], str(out)))
