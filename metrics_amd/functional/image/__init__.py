from metrics_amd.functional.image.misc import (
    error_relative_global_dimensionless_synthesis,
    relative_average_spectral_error,
    root_mean_squared_error_using_sliding_window,
    spatial_correlation_coefficient,
    spectral_angle_mapper,
    total_variation,
    universal_image_quality_index,
    visual_information_fidelity,
)
from metrics_amd.functional.image.pansharpening import (
    quality_with_no_reference,
    spatial_distortion_index,
    spectral_distortion_index,
)
from metrics_amd.functional.image.psnr import (
    peak_signal_noise_ratio,
    peak_signal_noise_ratio_with_blocked_effect,
)
from metrics_amd.functional.image.ssim import (
    multiscale_structural_similarity_index_measure,
    structural_similarity_index_measure,
)
from metrics_amd.functional.image.gradients import image_gradients
from metrics_amd.functional.image.perceptual import (
    learned_perceptual_image_patch_similarity,
    perceptual_path_length,
)
