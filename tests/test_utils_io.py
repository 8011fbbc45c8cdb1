"""Logger / scalar-writer / image-writer round trips (reference
misc/utils.py:118-252 capabilities)."""
import json
import os

import torch


def test_scalar_writer_jsonl_roundtrip(tmp_path):
    from p2pvg_amd.utils import ScalarWriter

    w = ScalarWriter(str(tmp_path))
    w.add_scalar("Train/mse", 0.5, 1)
    w.add_scalar("Train/mse", torch.tensor(0.25), 2)
    w.add_scalar("Eval/ssim", 0.9, 1)
    w.close()
    files = [f for f in os.listdir(tmp_path) if f.endswith(".jsonl")]
    assert files, "no scalar jsonl written"
    rows = [json.loads(l) for f in files
            for l in open(os.path.join(tmp_path, f)) if l.strip()]
    mse = [r for r in rows if r["tag"] == "Train/mse"]
    assert [r["step"] for r in mse] == [1, 2]
    assert abs(mse[1]["v"] - 0.25) < 1e-9


def test_save_image_and_gif_roundtrip(tmp_path):
    from PIL import Image

    from p2pvg_amd.utils.image import make_grid, save_gif, save_image, to_uint8_hwc

    frames = [torch.rand(3, 16, 16) for _ in range(4)]
    grid = make_grid(torch.stack(frames), nrow=2)
    png = str(tmp_path / "grid.png")
    save_image(grid, png)
    im = Image.open(png)
    assert im.size[0] >= 16 and im.size[1] >= 16

    gif = str(tmp_path / "clip.gif")
    save_gif(gif, [to_uint8_hwc(f) for f in frames])
    g = Image.open(gif)
    assert getattr(g, "n_frames", 1) == 4


def test_store_cmd_writes_argv(tmp_path):
    from p2pvg_amd.utils import store_cmd

    store_cmd(str(tmp_path))
    found = False
    for root, _, files in os.walk(tmp_path):
        for f in files:
            txt = open(os.path.join(root, f)).read()
            if "python" in txt or "pytest" in txt or len(txt) > 0:
                found = True
    assert found, "store_cmd wrote nothing"


def test_store_cmd_snapshots_entry_script(tmp_path):
    """store_cmd copies the invoking script's source into <log_dir>/src
    (reference misc/utils.py:227-229 provenance dump)."""
    import sys

    from p2pvg_amd.utils import store_cmd

    script = tmp_path / "fake_train.py"
    script.write_text("print('hi')\n")
    argv0 = sys.argv[0]
    sys.argv[0] = str(script)
    try:
        store_cmd(str(tmp_path / "log"))
    finally:
        sys.argv[0] = argv0
    snap = tmp_path / "log" / "src" / "fake_train.py"
    assert snap.exists() and snap.read_text() == "print('hi')\n"


def test_get_logger_writes_file(tmp_path):
    from p2pvg_amd.utils import get_logger

    path = str(tmp_path / "logs")
    logger = get_logger(path)
    logger.info("hello-round-1")
    for h in list(logger.handlers):
        h.flush()
    assert "hello-round-1" in open(path).read()
