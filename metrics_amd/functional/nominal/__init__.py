from metrics_amd.functional.nominal.metrics import (
    cramers_v,
    cramers_v_matrix,
    pearsons_contingency_coefficient_matrix,
    theils_u_matrix,
    tschuprows_t_matrix,
    fleiss_kappa,
    pearsons_contingency_coefficient,
    theils_u,
    tschuprows_t,
)
