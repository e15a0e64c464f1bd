"""The in-tree gfx950 extension must be rebuilt whenever kernel sources
change — a stale melgan_amd/_C.so silently runs OLD kernels on the GPU
box (the .so travels with the snapshot; sources do not get rebuilt
there).  This check makes staleness loud in CPU CI."""


def test_extension_is_current():
    from melgan_amd.kernels.build import needs_rebuild

    assert not needs_rebuild(), (
        "melgan_amd/_C.so is older than the kernel sources — run "
        "`python -m melgan_amd.kernels.build` and commit/ship the "
        "rebuilt extension")
