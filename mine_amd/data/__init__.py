from mine_amd.data.synthetic import SyntheticMPIDataset, collate_src_tgt  # noqa: F401


def get_dataset(config, logger=None, is_validation: bool = False):
    """Dataset factory (ref train.py:69-103).

    "llff" -> COLMAP-backed NeRFDataset; "synthetic" ->
    SyntheticMPIDataset of the same item schema. The remaining reference
    dataset names (realestate10k, flowers, kitti_raw, dtu) raise
    NotImplementedError exactly like the reference (ref train.py:100-101)
    — a user pointing them at real data must not silently train on
    noise. Benchmarks that want the synthetic generator under one of
    those names opt in with ``data.allow_synthetic_fallback: true``
    (bench.py builds SyntheticMPIDataset directly).
    """
    name = config["data.name"]
    known = ("llff", "realestate10k", "flowers", "kitti_raw", "dtu", "synthetic")
    assert name in known, name

    if name == "llff":
        from mine_amd.data.llff import NeRFDataset
        root = config["data.training_set_path"]
        return NeRFDataset(config, logger, root=root, is_validation=is_validation,
                           img_size=(config["data.img_w"], config["data.img_h"]),
                           supervision_count=config["data.num_tgt_views"],
                           visible_points_count=config["data.visible_point_count"],
                           img_pre_downsample_ratio=config["data.img_pre_downsample_ratio"])
    if name == "flowers":
        import os
        root = config["data.training_set_path"]
        if isinstance(root, str) and \
                os.path.exists(os.path.join(root, "cam_params.txt")):
            from mine_amd.data.flowers import FlowersDataset
            return FlowersDataset(config, logger, root=root,
                                  is_validation=is_validation)
    if name != "synthetic" and not bool(
            config.get("data.allow_synthetic_fallback", False)):
        raise NotImplementedError(
            f"dataset pipeline '{name}' is not implemented (the reference "
            "raises here too, ref train.py:100-101); set data.name: "
            "synthetic or data.allow_synthetic_fallback: true to train on "
            "the synthetic generator")
    if logger is not None and name != "synthetic":
        logger.warning("dataset '%s': using the SYNTHETIC generator "
                       "(data.allow_synthetic_fallback is set)", name)
    return SyntheticMPIDataset(config, is_validation=is_validation)
