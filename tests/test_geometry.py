import torch

from mine_amd.utils.geometry import inverse_3x3, inverse_4x4, inverse_rigid_4x4


def test_inverse_3x3_matches_torch():
    torch.manual_seed(0)
    m = torch.randn(64, 3, 3, dtype=torch.float64)
    m += 3 * torch.eye(3, dtype=torch.float64)  # keep well-conditioned
    inv = inverse_3x3(m)
    torch.testing.assert_close(inv, torch.inverse(m), rtol=1e-9, atol=1e-9)


def test_inverse_3x3_intrinsics():
    K = torch.tensor([[[500.0, 0.0, 192.0], [0.0, 500.0, 128.0], [0.0, 0.0, 1.0]]])
    Kinv = inverse_3x3(K)
    torch.testing.assert_close(K @ Kinv, torch.eye(3).unsqueeze(0),
                               rtol=1e-5, atol=1e-5)


def test_inverse_rigid_4x4():
    torch.manual_seed(1)
    # random rotation via QR
    q, _ = torch.linalg.qr(torch.randn(8, 3, 3, dtype=torch.float64))
    det = torch.det(q).unsqueeze(-1).unsqueeze(-1)
    q = q * det.sign()
    G = torch.zeros(8, 4, 4, dtype=torch.float64)
    G[:, :3, :3] = q
    G[:, :3, 3] = torch.randn(8, 3, dtype=torch.float64)
    G[:, 3, 3] = 1.0
    torch.testing.assert_close(inverse_rigid_4x4(G) @ G,
                               torch.eye(4, dtype=torch.float64).expand(8, 4, 4),
                               rtol=1e-9, atol=1e-9)


def test_inverse_4x4_general():
    torch.manual_seed(2)
    m = torch.randn(16, 4, 4, dtype=torch.float64) + 4 * torch.eye(4, dtype=torch.float64)
    torch.testing.assert_close(inverse_4x4(m), torch.inverse(m), rtol=1e-8, atol=1e-8)


def test_batch_size_two_no_nan():
    # the exact failure mode the reference worked around (pytorch#47272,
    # ref utils.py:96-117): batched inverse at B=2
    m = torch.eye(3).unsqueeze(0).repeat(2, 1, 1)
    assert not torch.isnan(inverse_3x3(m)).any()
