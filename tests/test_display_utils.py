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
