"""WatDiv-shaped star/linear/snowflake templates (BASELINE.json
configs[3]); schema ids match csrc/watdiv_gen.cpp.  Constants are
closed-form hub ids (genre g = GENRE0+g, retailer r = RETAILER0+r)."""
from . import Plan, DIR_IN, DIR_OUT

(HASGENRE, OFFER_PRODUCT, RETAILER, REVIEW_PRODUCT, REVIEWER, PURCHASED,
 FRIEND) = range(2, 9)
(T_PRODUCT, T_OFFER, T_REVIEW, T_USER, T_GENRE, T_RETAILER) = range(9, 15)
TYPE_ID = 1
GENRE0 = 1 << 17
RETAILER0 = GENRE0 + 250

P, O, R, REV, U, F = -1, -2, -3, -4, -5, -6


def star(genre=GENRE0):
    """S: products of a genre, their offers, the offers' retailers."""
    return Plan([
        (genre, HASGENRE, DIR_IN, P),
        (P, TYPE_ID, DIR_OUT, T_PRODUCT),
        (P, OFFER_PRODUCT, DIR_IN, O),
        (O, RETAILER, DIR_OUT, R),
    ], nvars=3, required_vars=[P, O, R])


def linear(genre=GENRE0):
    """L: genre -> products -> reviews -> reviewers -> their friends."""
    return Plan([
        (genre, HASGENRE, DIR_IN, P),
        (P, REVIEW_PRODUCT, DIR_IN, REV),
        (REV, REVIEWER, DIR_OUT, U),
        (U, FRIEND, DIR_OUT, F),
    ], nvars=6, required_vars=[P, REV, U, F])


def snowflake(genre=GENRE0):
    """F: genre-rooted two-armed join (offers+retailer, reviews+user)."""
    return Plan([
        (genre, HASGENRE, DIR_IN, P),
        (P, OFFER_PRODUCT, DIR_IN, O),
        (O, RETAILER, DIR_OUT, R),
        (P, REVIEW_PRODUCT, DIR_IN, REV),
        (REV, REVIEWER, DIR_OUT, U),
    ], nvars=5, required_vars=[P, O, R, REV, U])


def retailer_star(retailer=RETAILER0):
    """S2: retailer hub (in-degree ~10N/1000) -> offers -> products."""
    return Plan([
        (retailer, RETAILER, DIR_IN, O),
        (O, TYPE_ID, DIR_OUT, T_OFFER),
        (O, OFFER_PRODUCT, DIR_OUT, P),
        (P, TYPE_ID, DIR_OUT, T_PRODUCT),
    ], nvars=3, required_vars=[O, P])


def purchase_chain(genre=GENRE0):
    """L2: genre -> products -> purchasers -> friends -> their purchases
    (k2u hub stress: the last hop re-expands)."""
    return Plan([
        (genre, HASGENRE, DIR_IN, P),
        (P, PURCHASED, DIR_IN, U),
        (U, FRIEND, DIR_OUT, F),
        (F, PURCHASED, DIR_OUT, -6 - 1),
    ], nvars=7, required_vars=[P, U, F, -7])


ALL = {"w_star": star(), "w_linear": linear(), "w_snow": snowflake(),
       "w_rstar": retailer_star(), "w_chain": purchase_chain()}
