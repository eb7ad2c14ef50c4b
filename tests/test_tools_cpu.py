"""tools/rocpd_stats.py works against a synthetic rocpd database, so the
profiling evidence pipeline is itself tested."""
import sqlite3

from tools.rocpd_stats import kernel_stats, pmc_stats

GUID = "_test"


def _mk_db(path):
    db = sqlite3.connect(path)
    db.execute(f"CREATE TABLE rocpd_kernel_dispatch{GUID} "
               "(id, kernel_id, event_id, start, end)")
    db.execute(f"CREATE TABLE rocpd_info_kernel_symbol{GUID} "
               "(id, display_name)")
    db.execute(f"CREATE TABLE rocpd_info_pmc{GUID} (id, name)")
    db.execute(f"CREATE TABLE rocpd_pmc_event{GUID} (event_id, pmc_id, value)")
    db.execute(f"INSERT INTO rocpd_info_kernel_symbol{GUID} VALUES (1, 'fast_kernel')")
    db.execute(f"INSERT INTO rocpd_info_kernel_symbol{GUID} VALUES (2, 'slow_kernel')")
    # fast: 2 calls x 1000 ns; slow: 1 call x 8000 ns
    db.execute(f"INSERT INTO rocpd_kernel_dispatch{GUID} VALUES (1, 1, 10, 0, 1000)")
    db.execute(f"INSERT INTO rocpd_kernel_dispatch{GUID} VALUES (2, 1, 11, 2000, 3000)")
    db.execute(f"INSERT INTO rocpd_kernel_dispatch{GUID} VALUES (3, 2, 12, 0, 8000)")
    db.execute(f"INSERT INTO rocpd_info_pmc{GUID} VALUES (100, 'SQ_VALU_MFMA_BUSY_CYCLES')")
    db.execute(f"INSERT INTO rocpd_info_pmc{GUID} VALUES (101, 'SQ_WAVE_CYCLES')")
    db.execute(f"INSERT INTO rocpd_info_pmc{GUID} VALUES (102, 'SQ_WAIT_ANY')")
    for eid, mfma, wave, wait in [(10, 400, 1000, 100), (11, 400, 1000, 100),
                                  (12, 0, 8000, 4000)]:
        db.execute(f"INSERT INTO rocpd_pmc_event{GUID} VALUES ({eid}, 100, {mfma})")
        db.execute(f"INSERT INTO rocpd_pmc_event{GUID} VALUES ({eid}, 101, {wave})")
        db.execute(f"INSERT INTO rocpd_pmc_event{GUID} VALUES ({eid}, 102, {wait})")
    db.commit()
    db.close()


def test_kernel_stats(tmp_path):
    p = str(tmp_path / "t.db")
    _mk_db(p)
    out = kernel_stats(p)
    lines = out.splitlines()
    # slow_kernel dominates (80% of 10 us total) and sorts first
    assert "slow_kernel" in lines[1]
    assert "80.00" in lines[1]
    assert "fast_kernel" in lines[2]
    assert "total GPU kernel time: 0.010 ms over 3 dispatches" in out


def test_pmc_stats(tmp_path):
    p = str(tmp_path / "t.db")
    _mk_db(p)
    out = pmc_stats(p)
    # generic reporter: every captured counter appears as a raw sum column
    fast = [l for l in out.splitlines() if "fast_kernel" in l][0]
    assert "2.000e+03" in fast and "8.000e+02" in fast  # wave cycles, mfma
    slow = [l for l in out.splitlines() if "slow_kernel" in l][0]
    assert "8.000e+03" in slow and "4.000e+03" in slow  # waves, wait
