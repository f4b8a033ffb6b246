from metrics_amd.functional.nominal.metrics import (
    cramers_v,
    fleiss_kappa,
    pearsons_contingency_coefficient,
    theils_u,
    tschuprows_t,
)
