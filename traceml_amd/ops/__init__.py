"""Native HIP extensions (gfx950). Built in-tree by build_ext so the .so
travels with the repo snapshot to GPU boxes."""
