from .metrics import mse, l1, rmse, psnr, ssim  # noqa: F401
from .lpips import PerceptualLoss, LPIPS  # noqa: F401
