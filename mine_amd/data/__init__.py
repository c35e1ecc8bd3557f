from mine_amd.data.synthetic import SyntheticMPIDataset, collate_src_tgt  # noqa: F401


def get_dataset(config, logger=None, is_validation: bool = False):
    """Dataset factory (ref train.py:69-103).

    "llff" -> COLMAP-backed NeRFDataset; "synthetic" (and any benchmark
    run without data on disk) -> SyntheticMPIDataset of the same item
    schema. The remaining reference dataset names (realestate10k,
    flowers, kitti_raw, dtu) use the synthetic generator when their
    training_set_path does not exist — the reference never shipped those
    pipelines either (ref train.py:100-101 raises NotImplementedError
    for everything but llff).
    """
    import os

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
    return SyntheticMPIDataset(config, is_validation=is_validation)
