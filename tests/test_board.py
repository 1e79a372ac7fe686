"""TensorBoard-on-Volume role: event logging, parsing, wsgi dashboard."""
import modal_examples_amd as modal
from modal_examples_amd.observability.board import (
    VolumeReloadMiddleware,
    log_scalar,
    make_board_wsgi,
    read_runs,
)


def test_log_and_read(tmp_path):
    for s in range(5):
        log_scalar(tmp_path / "runA", "loss", s, 1.0 / (s + 1))
    log_scalar(tmp_path / "runB", "acc", 0, 0.5)
    runs = read_runs(tmp_path)
    assert set(runs) == {"runA", "runB"}
    assert runs["runA"]["loss"][0] == (0, 1.0) and len(runs["runA"]["loss"]) == 5


def test_wsgi_board_serves_html_and_json(tmp_path):
    import json

    for s in range(3):
        log_scalar(tmp_path / "r", "train/loss", s, 2.0 - s * 0.5)

    vol = modal.Volume.from_name("board-test-vol", create_if_missing=True)
    app = VolumeReloadMiddleware(make_board_wsgi(tmp_path), vol)

    def call(path):
        out = {}

        def sr(status, headers):
            out["status"] = status

        body = b"".join(app({"PATH_INFO": path, "REQUEST_METHOD": "GET"}, sr))
        return out["status"], body

    st, body = call("/data")
    assert st.startswith("200")
    assert json.loads(body)["r"]["train/loss"][2] == [2, 1.0]
    st, body = call("/")
    assert b"<svg" in body and b"train/loss" in body
