"""Display layout math (no X server required)."""

from selkies_amd import display_utils as D


def test_align_dims_16():
    assert D.align_dims_16(1920, 1080) == (1920, 1072)
    assert D.align_dims_16(17, 15) == (16, 16)
    assert D.align_dims_16(0, 5) == (16, 16)


def test_cvt_modeline_sane():
    name, mode = D.cvt_modeline(1920, 1080, 60)
    assert name == "1920x1080_60.00"
    parts = mode.split()
    pclk = float(parts[0])
    htotal, vtotal = int(parts[4]), int(parts[8])
    # refresh from the modeline lands near 60 Hz
    refresh = pclk * 1e6 / (htotal * vtotal)
    assert 59 < refresh < 61
    assert htotal > 1920 and vtotal > 1080


def test_dual_layout_positions():
    mons = D.compute_dual_layout(1920, 1080, 1280, 720, "right")
    assert mons[0].x == 0 and mons[1].x == 1920
    w, h = D.framebuffer_bounds(mons)
    assert (w, h) == (1920 + 1280, 1072)

    mons = D.compute_dual_layout(1920, 1080, 1280, 720, "left")
    w, h = D.framebuffer_bounds(mons)
    assert (w, h) == (1920 + 1280, 1072)
    assert mons[0].x == 1280 and mons[1].x == 0

    mons = D.compute_dual_layout(1920, 1080, 1920, 1080, "below")
    w, h = D.framebuffer_bounds(mons)
    assert (w, h) == (1920, 2144)

    mons = D.compute_dual_layout(800, 600, 800, 600, "above")
    w, h = D.framebuffer_bounds(mons)
    assert (w, h) == (800, 1184)
    assert mons[0].y == 592 and mons[1].y == 0  # normalized origins


def test_parse_dri_node():
    assert D.parse_dri_node_to_index("/dev/dri/renderD128") == 0
    assert D.parse_dri_node_to_index("/dev/dri/renderD129") == 1
    assert D.parse_dri_node_to_index("bogus") == -1


def test_monitor_geometry_and_cursor_scale():
    from selkies_amd import display_utils as du
    # 1920x1080 at 96 dpi -> 508x286 mm
    assert du.monitor_geometry(1920, 1080, 0, 0, 96) == \
        "1920/508x1080/286+0+0"
    g = du.monitor_geometry(2560, 1440, 1920, 0, 192)
    assert g.endswith("+1920+0")
    assert "2560/339" in g
    assert du.cursor_size_for_dpi(96) == 24
    assert du.cursor_size_for_dpi(192) == 48
    assert du.cursor_size_for_dpi(120, 32) == 40


def test_logical_monitor_commands(monkeypatch):
    from selkies_amd import display_utils as du
    calls = []

    def fake_run(cmd, display):
        calls.append(cmd)
        return ""
    monkeypatch.setattr(du, "_run", fake_run)
    assert du.set_logical_monitor("selkies-0", 1920, 1080, 0, 0)
    assert calls[-1][:3] == ["xrandr", "--setmonitor", "selkies-0"]
    assert du.delete_logical_monitor("selkies-0")
    assert calls[-1] == ["xrandr", "--delmonitor", "selkies-0"]
    assert du.apply_logical_dual_layout(1920, 1080, 1280, 1024)
    fbs = [c for c in calls if c[1] == "--fb"]
    assert fbs and fbs[-1][2] == "3200x1072"  # heights 16-aligned (1080->1072)
    mons = [c for c in calls if c[1] == "--setmonitor"]
    assert len(mons) == 3


def test_xresources_dpi_persist(tmp_path):
    from selkies_amd import display_utils as du
    p = tmp_path / "Xresources"
    p.write_text("Xft.antialias: 1\nXft.dpi: 96\nXcursor.size: 24\n")
    assert du._persist_xresources_dpi(144, str(p))
    txt = p.read_text()
    assert "Xft.dpi:   144" in txt
    assert txt.count("Xft.dpi") == 1
    assert "Xft.antialias: 1" in txt
    assert "Xcursor.size: 24" in txt
