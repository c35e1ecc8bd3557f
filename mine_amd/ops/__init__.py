"""Differentiable MPI rendering ops.

Public API (device-dispatched):
    render_src_view   — volume-composite the MPI in the source camera,
                        optionally RGB-blended with the real source image
                        (ref operations/mpi_rendering.py:42-82 +
                        synthesis_task.py:267-274, fused)
    render_tgt_view   — plane-sweep homography warp + z-cull + composite
                        into a novel view (ref operations/
                        homography_sampler.py:58-141 +
                        mpi_rendering.py:181-241, fused)
    sample_disparity* — stratified disparity sampling
    sample_pdf        — inverse-CDF hierarchical sampling
    gather_pixel_by_pxpy, ssim, edge-aware losses, psnr

On CUDA (ROCm) devices these run hand-written CDNA4 HIP kernels; on CPU
they run the pure-torch reference implementations in
`mine_amd.ops.torch_ref` (which double as the numerics oracle in tests).
"""
from mine_amd.ops.torch_ref import (  # noqa: F401
    alpha_composite,
    gather_pixel_by_pxpy,
    make_meshgrid,
    sample_disparity_from_bins,
    sample_disparity_linspace,
    sample_pdf,
    src_plane_xyz,
    tgt_plane_xyz,
    homography_tgt_to_src,
    volume_composite,
    weighted_sum_mpi,
)
from mine_amd.ops.renderer import render_src_view, render_tgt_view  # noqa: F401
from mine_amd.ops.ssim import ssim  # noqa: F401
from mine_amd.ops.losses import (  # noqa: F401
    edge_aware_loss,
    edge_aware_loss_v2,
    psnr,
)
