from metrics_amd.functional.audio.metrics import (
    complex_scale_invariant_signal_noise_ratio,
    permutation_invariant_training,
    pit_permutate,
    scale_invariant_signal_distortion_ratio,
    scale_invariant_signal_noise_ratio,
    signal_distortion_ratio,
    signal_noise_ratio,
    source_aggregated_signal_distortion_ratio,
)
from metrics_amd.functional.audio.external import (
    deep_noise_suppression_mean_opinion_score,
    non_intrusive_speech_quality_assessment,
    perceptual_evaluation_speech_quality,
    short_time_objective_intelligibility,
    speech_reverberation_modulation_energy_ratio,
)
