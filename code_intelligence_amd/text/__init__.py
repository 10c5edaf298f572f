from .tokenizer import (Tokenizer, Vocab, defaults_specials, fix_html,
                        markdown_rules, replace_rep, replace_wrep,
                        spec_add_spaces, rm_useless_spaces, replace_all_caps,
                        deal_caps, UNK, PAD, BOS, EOS, FLD, TK_MAJ, TK_UP,
                        TK_REP, TK_WREP)

__all__ = [
    "Tokenizer", "Vocab", "defaults_specials", "fix_html", "markdown_rules",
    "replace_rep", "replace_wrep", "spec_add_spaces", "rm_useless_spaces",
    "replace_all_caps", "deal_caps", "UNK", "PAD", "BOS", "EOS", "FLD",
    "TK_MAJ", "TK_UP", "TK_REP", "TK_WREP",
]
