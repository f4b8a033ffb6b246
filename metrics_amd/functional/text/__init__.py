from metrics_amd.functional.text.bleu import bleu_score, chrf_score, sacre_bleu_score, translation_edit_rate
from metrics_amd.functional.text.error_rates import (
    char_error_rate,
    edit_distance,
    match_error_rate,
    word_error_rate,
    word_information_lost,
    word_information_preserved,
)
from metrics_amd.functional.text.misc import extended_edit_distance, perplexity, squad
from metrics_amd.functional.text.rouge import rouge_score
from metrics_amd.functional.text.bert_infolm import bert_score, infolm
