"""Unit test for the rocprofv3 db summarizer (profiling subsystem)."""

import sqlite3

from k3samd.utils.rocpd_summary import summarize, shorten


def make_db(path):
    con = sqlite3.connect(path)
    con.execute("CREATE TABLE top_kernels "
                "(name TEXT, total_calls INT, total_duration REAL, "
                "average REAL, percentage REAL)")
    con.executemany(
        "INSERT INTO top_kernels VALUES (?,?,?,?,?)",
        [("void k3samd_kern::stream_triad_kernel<true>(float4*, ...)",
          40, 20616.2, 515.4, 97.6),
         ("void at::native::some_huge_template<with, args>(x)",
          2, 503.9, 252.0, 2.4)])
    con.execute("CREATE TABLE rocpd_info_agent (type TEXT, name TEXT)")
    con.execute("INSERT INTO rocpd_info_agent VALUES ('GPU', 'gfx950')")
    con.commit()
    con.close()


def test_summarize(tmp_path):
    db = tmp_path / "x_results.db"
    make_db(db)
    md = summarize(db)
    assert "stream_triad_kernel" in md
    assert "97.6" in md
    assert "gfx950" in md
    assert md.count("|") > 10  # table rendered


def test_shorten():
    assert shorten("void foo(int, float)") == "void foo"
    long = "x" * 300 + "(args)"
    assert len(shorten(long)) <= 100
