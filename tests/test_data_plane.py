"""Data-plane tests (mirror reference test_data_server.py: file-list
slicing + meta balance over a real server on 127.0.0.1)."""
from edl_amd.data.data_server import DataClient, DataServer, PodsData
from edl_amd.data.dataset import TxtFileSplitter
from edl_amd.data.reader import Reader


def make_files(tmp_path, n_files=4, lines_per=6):
    files = []
    for i in range(n_files):
        p = tmp_path / ("f%d.txt" % i)
        p.write_text("".join("file%d-line%d\n" % (i, j) for j in range(lines_per)))
        files.append(str(p))
    return files


def test_file_slicing_round_robin():
    pd = PodsData(["a", "b", "c", "d", "e"], ["p0", "p1"])
    assert pd.get_file_list("p0") == ["a", "c", "e"]
    assert pd.get_file_list("p1") == ["b", "d"]
    assert pd.get_file_list("ghost") == []


def test_balance_steals_from_fast_pod():
    pd = PodsData([], ["p0", "p1"])
    pd.report("p0", ["b%d" % i for i in range(8)])
    pd.report("p1", [])  # slow pod reported nothing
    a0, _ = pd.take_assignments("p0")
    a1, _ = pd.take_assignments("p1")
    assert len(a0) + len(a1) == 8
    assert len(a1) >= 3  # leveled to ~average, not starving

def test_balance_done_after_all_finish():
    pd = PodsData([], ["p0"])
    pd.report("p0", ["b0", "b1"], finished=True)
    items, done = pd.take_assignments("p0")
    assert [b for _, b in items] == ["b0", "b1"]
    assert done


def test_txt_splitter(tmp_path):
    p = tmp_path / "x.txt"
    p.write_text("a\nb\n\nc\n")
    recs = list(TxtFileSplitter()(str(p)))
    assert recs == [(0, "a"), (1, "b"), (2, "c")]


def test_server_roundtrip(tmp_path):
    files = make_files(tmp_path, 2, 3)
    srv = DataServer(file_list=files, pod_ids=["p0"]).start()
    try:
        cli = DataClient("127.0.0.1:%d" % srv.port)
        assert cli.get_file_list("p0") == files
        cli.put_batch("b0", {"x": 1})
        assert cli.get_batch("b0") == {"x": 1}
        cli.report("p0", ["b0"], finished=True)
        items, done = cli.get_meta("p0")
        assert items == [("p0", "b0")] and done
        cli.close()
    finally:
        srv.stop()


def test_two_pod_reader_rebalance(tmp_path):
    """Two pods read a shared file set; every record is delivered exactly
    once across pods, with remote fetch for stolen batches."""
    files = make_files(tmp_path, 4, 6)  # 24 records total
    # leader = p0's server (knows the file list / pod ids)
    s0 = DataServer(file_list=files, pod_ids=["p0", "p1"]).start()
    s1 = DataServer().start()
    eps = {"p0": "127.0.0.1:%d" % s0.port, "p1": "127.0.0.1:%d" % s1.port}
    leader_ep = eps["p0"]
    try:
        r0 = Reader("p0", leader_ep, s0, eps, batch_size=2)
        r1 = Reader("p1", leader_ep, s1, eps, batch_size=2)
        import threading

        got = {"p0": [], "p1": []}

        def run(name, r):
            for item in r:
                got[name].extend(item["data"])

        t0 = threading.Thread(target=run, args=("p0", r0))
        t1 = threading.Thread(target=run, args=("p1", r1))
        t0.start(); t1.start()
        t0.join(30); t1.join(30)
        assert not t0.is_alive() and not t1.is_alive()
        all_recs = sorted(got["p0"] + got["p1"])
        expect = sorted("file%d-line%d" % (i, j) for i in range(4) for j in range(6))
        assert all_recs == expect  # exactly-once delivery
        assert got["p0"] and got["p1"]  # both pods participated
        r0.close(); r1.close()
    finally:
        s0.stop(); s1.stop()


def test_record_image_set_deterministic():
    """plane.RecordImageSet: equal records -> equal tensors on any rank
    (elastic resizes must resume on identical data); labels parse from a
    leading integer, else derive from the record hash."""
    import torch

    from edl_amd.data.plane import RecordImageSet

    recs = ["3 img-a", "7 img-b", "not-an-int payload", "3 img-a"]
    ds1 = RecordImageSet(list(recs), batch_size=2, device=torch.device("cpu"),
                         image_shape=(3, 8, 8), num_classes=10)
    ds2 = RecordImageSet(list(recs), batch_size=2, device=torch.device("cpu"),
                         image_shape=(3, 8, 8), num_classes=10)
    x1, y1 = ds1.next()
    x2, y2 = ds2.next()
    assert torch.equal(x1, x2) and torch.equal(y1, y2)
    assert y1.tolist() == [3, 7]
    xb, yb = ds1.next()
    assert 0 <= yb[0] < 10          # hash-derived label in range
    assert yb[1] == 3               # wraps to equal record -> equal label
    assert torch.equal(xb[1], x1[0])  # equal record -> equal image
    assert ds1.steps() == 2


def test_prof_summary_parses_rocpd_schema(tmp_path, capsys):
    """tools/prof_summary.py against a synthetic rocpd sqlite db (locks
    the rocprofv3 schema assumptions the docs cite)."""
    import sqlite3
    import sys

    sys.path.insert(0, str(__import__("pathlib").Path(__file__).parents[1]))
    from tools.prof_summary import main as summarize

    db = tmp_path / "x_results.db"
    con = sqlite3.connect(db)
    con.execute("CREATE TABLE rocpd_kernel_dispatch_abc "
                "(kernel_id INT, start INT, end INT)")
    con.execute("CREATE TABLE rocpd_info_kernel_symbol_abc "
                "(id INT, display_name TEXT)")
    con.execute("INSERT INTO rocpd_info_kernel_symbol_abc VALUES (1, 'k1')")
    con.execute("INSERT INTO rocpd_info_kernel_symbol_abc VALUES (2, 'k2')")
    for kid, st, en in [(1, 0, 1000), (1, 2000, 4000), (2, 0, 500)]:
        con.execute("INSERT INTO rocpd_kernel_dispatch_abc VALUES (?,?,?)",
                    (kid, st, en))
    con.commit()
    con.close()

    summarize(str(db))
    out = capsys.readouterr().out
    assert "total kernels: 3" in out
    assert out.index("k1") < out.index("k2")  # sorted by total time
