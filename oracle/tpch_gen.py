"""Seeded synthetic TPC-H data generator (numpy), TEST INFRASTRUCTURE ONLY.

Generates the tables/columns the hot path's queries touch (TPC-H Q1/Q3/Q5/Q6
per the reference's apps/tpc-h/tpch.py + tpch_ref.py), with TPC-H-spec
distributions so selectivities match the reference's workload:

- orders:   1,500,000 x SF rows; o_orderdate uniform [1992-01-01, 1998-08-02]
- lineitem: 1..7 lines per order (avg 4 -> ~6,000,000 x SF rows);
            l_shipdate = o_orderdate + U[1,121] days;
            l_receiptdate = l_shipdate + U[1,30];
            l_returnflag 'R'/'A' if receipt <= 1995-06-17 else 'N';
            l_linestatus 'O' if ship > 1995-06-17 else 'F';
            l_discount = k/100, k in U[0,10]; l_tax = k/100, k in U[0,8];
            l_quantity = U[1,50]; l_extendedprice = quantity * retailprice
            with the spec's p_retailprice formula over a synthetic partkey.
- customer: 150,000 x SF; c_mktsegment uniform over the spec's 5 segments.
- supplier: 10,000 x SF; s_nationkey uniform 0..24.
- nation/region: the spec's fixed 25/5 rows.

Key structure follows TPC-H spec 4.2.3: o_orderkey is SPARSE (8 keys per
32-key bucket, unique within [1, 4 x SF x 1,500,000] — dbgen's
sparse encoding), and o_custkey never satisfies custkey % 3 == 0 (the
spec's customer-mortality hole: a third of customers never order).

Deviation from dbgen (documented): comment/text columns are absent, and
the RNG streams are numpy Philox, not dbgen's per-column Lehmer streams —
dbgen's per-stream seed table lives only in dbgen's source (rnd.c), which
is neither vendored in /root/reference nor fetchable here, so row-level
bit-compatibility with dbgen (and therefore assertion against the TPC-H
published answer sets) is out of reach in this environment. The external
anchors used instead: (a) the spec 4.2 distributions above, (b) pyarrow
Acero (an independent C++ join/group-by engine) restatements of each
query asserted equal to this oracle on the same inputs
(tests/test_oracle_acero.py), (c) pyarrow's own decoders for the file
formats. Parity GPU-vs-oracle runs on the SAME generated inputs
(DESIGN.md §Oracle).

Dates are encoded as int32 days since 1970-01-01 (Arrow date32).
Low-cardinality strings are dictionary codes (u8) + the code tables below.

All randomness from numpy's Philox via default_rng(seed): fully deterministic
across platforms for a given (seed, sf).
"""
import datetime
import numpy as np

EPOCH = datetime.date(1970, 1, 1)


def days(y, m, d):
    """date32 value (days since 1970-01-01) of a calendar date."""
    return (datetime.date(y, m, d) - EPOCH).days


# dictionary code tables (sorted, so code order == lexicographic order)
RETURNFLAG = ["A", "N", "R"]          # l_returnflag codes 0,1,2
LINESTATUS = ["F", "O"]               # l_linestatus codes 0,1
MKTSEGMENT = ["AUTOMOBILE", "BUILDING", "FURNITURE", "HOUSEHOLD", "MACHINERY"]
ORDERPRIORITY = ["1-URGENT", "2-HIGH", "3-MEDIUM", "4-NOT SPECIFIED",
                 "5-LOW"]             # spec 4.2.3 uniform, sorted order
SHIPMODE = ["AIR", "FOB", "MAIL", "RAIL", "REG AIR", "SHIP", "TRUCK"]
SHIPINSTRUCT = ["COLLECT COD", "DELIVER IN PERSON", "NONE",
                "TAKE BACK RETURN"]
# p_container = size x kind (spec 4.2.2.13); code = c1 * 8 + c2
CONTAINER1 = ["JUMBO", "LG", "MED", "SM", "WRAP"]
CONTAINER2 = ["BAG", "BOX", "CAN", "CASE", "DRUM", "JAR", "PACK", "PKG"]
P_NAME_FOREST = 29       # code of "forest" in the 92-word p_name space


def container_code(name):
    a, b = name.split(" ", 1)
    return CONTAINER1.index(a) * 8 + CONTAINER2.index(b)


def brand_code(name):
    mn = name.split("#", 1)[1]           # 'Brand#MN', M,N in 1..5
    return (int(mn[0]) - 1) * 5 + (int(mn[1]) - 1)
# p_type = syllable1 x syllable2 x syllable3 (spec 4.2.2.13); code =
# ((s1 * 5) + s2) * 5 + s3; LIKE 'PROMO%' == s1 == index of "PROMO"
PTYPE_SYL1 = ["ECONOMY", "LARGE", "MEDIUM", "PROMO", "SMALL", "STANDARD"]
PTYPE_SYL2 = ["ANODIZED", "BRUSHED", "BURNISHED", "PLATED", "POLISHED"]
PTYPE_SYL3 = ["BRASS", "COPPER", "NICKEL", "STEEL", "TIN"]
PTYPE_PROMO_SYL1 = PTYPE_SYL1.index("PROMO")

# TPC-H spec nation table: (name, regionkey)
NATIONS = [
    ("ALGERIA", 0), ("ARGENTINA", 1), ("BRAZIL", 1), ("CANADA", 1),
    ("EGYPT", 4), ("ETHIOPIA", 0), ("FRANCE", 3), ("GERMANY", 3),
    ("INDIA", 2), ("INDONESIA", 2), ("IRAN", 4), ("IRAQ", 4),
    ("JAPAN", 2), ("JORDAN", 4), ("KENYA", 0), ("MOROCCO", 0),
    ("MOZAMBIQUE", 0), ("PERU", 1), ("CHINA", 2), ("ROMANIA", 3),
    ("SAUDI ARABIA", 4), ("VIETNAM", 2), ("RUSSIA", 3),
    ("UNITED KINGDOM", 3), ("UNITED STATES", 1),
]
REGIONS = ["AFRICA", "AMERICA", "ASIA", "EUROPE", "MIDDLE EAST"]

ORDERDATE_LO = days(1992, 1, 1)
ORDERDATE_HI = days(1998, 8, 2)       # 1998-12-31 - 151 days, inclusive
RECEIPT_CUTOFF = days(1995, 6, 17)

Q1_CUTOFF = days(1998, 9, 2)          # date '1998-12-01' - interval '90' day
Q3_DATE = days(1995, 3, 15)
Q5_LO = days(1994, 1, 1)
Q5_HI = days(1995, 1, 1)


def n_orders(sf):
    return int(round(1_500_000 * sf))


def n_customers(sf):
    return int(round(150_000 * sf))


def n_suppliers(sf):
    return int(round(10_000 * sf))


def n_parts(sf):
    return max(1, int(round(200_000 * sf)))


def _retailprice(partkey):
    """p_retailprice per TPC-H spec 4.2.3: (90000 + (pk/10)%20001 + 100*(pk%1000))/100."""
    pk = partkey.astype(np.int64)
    cents = 90000 + (pk // 10) % 20001 + 100 * (pk % 1000)
    return cents.astype(np.float64) / 100.0


def sparse_orderkeys(n):
    """TPC-H spec 4.2.3 sparse o_orderkey: 8 keys used per 32-key bucket,
    unique within [1, 4n] (dbgen's sparse encoding)."""
    idx = np.arange(n, dtype=np.int64)
    return (idx // 8) * 32 + (idx % 8) + 1


def gen_orders(sf, seed=42):
    """orders columns: o_orderkey i64 (sparse per spec 4.2.3), o_custkey
    i64 (custkey % 3 != 0 hole), o_orderdate i32 (date32), o_shippriority
    i32 (always 0)."""
    n = n_orders(sf)
    rng = np.random.default_rng([seed, 1])
    ck = rng.integers(1, n_customers(sf) + 1, n, dtype=np.int64)
    # spec: O_CUSTKEY must never be a multiple of 3 (customer mortality);
    # step multiples down one (c-1 >= 2 and (c-1) % 3 == 2)
    ck = np.where(ck % 3 == 0, ck - 1, ck)
    return {
        "o_orderkey": sparse_orderkeys(n),
        "o_custkey": ck,
        "o_orderdate": rng.integers(ORDERDATE_LO, ORDERDATE_HI + 1, n).astype(np.int32),
        "o_shippriority": np.zeros(n, dtype=np.int32),
        # spec 4.2.3: O_ORDERPRIORITY random over the 5 priorities (u8
        # codes into ORDERPRIORITY). O_TOTALPRICE is derived from the
        # order's lines; gen_lineitem fills it in when given this dict
        # (zeros until then).
        "o_orderpriority": rng.integers(0, 5, n).astype(np.uint8),
        "o_totalprice": np.zeros(n, dtype=np.float64),
        # text-boundary adaptation (header): the Q13 predicate
        # o_comment NOT LIKE '%special%requests%' becomes this flag
        # (dbgen plants the phrase in a small fraction of comments);
        # drawn after the columns above (stream-append stable).
        # o_orderstatus is DERIVED from the order's line statuses
        # (spec 4.2.3: F if all lines F, O if all O, else P) —
        # gen_lineitem fills it in (2 = P until then).
        "o_comment_special": (rng.random(n) < 0.019).astype(np.uint8),
        "o_orderstatus": np.full(n, 2, dtype=np.uint8),
    }


def gen_lineitem(sf, seed=42, orders=None):
    """lineitem columns (codes for flags): l_orderkey i64, l_suppkey i64,
    l_quantity f64, l_extendedprice f64, l_discount f64, l_tax f64,
    l_returnflag u8, l_linestatus u8, l_shipdate i32."""
    if orders is None:
        orders = gen_orders(sf, seed)
    rng = np.random.default_rng([seed, 2])
    no = len(orders["o_orderkey"])
    lines_per_order = rng.integers(1, 8, no)          # U[1,7]
    n = int(lines_per_order.sum())
    l_orderkey = np.repeat(orders["o_orderkey"], lines_per_order)
    o_date_rep = np.repeat(orders["o_orderdate"], lines_per_order).astype(np.int64)

    quantity = rng.integers(1, 51, n).astype(np.float64)
    partkey = rng.integers(1, n_parts(sf) + 1, n, dtype=np.int64)
    extendedprice = quantity * _retailprice(partkey)
    discount = rng.integers(0, 11, n).astype(np.float64) / 100.0
    tax = rng.integers(0, 9, n).astype(np.float64) / 100.0
    shipdate = o_date_rep + rng.integers(1, 122, n)
    commitdate = o_date_rep + rng.integers(30, 91, n)  # spec: +U[30,90]
    receiptdate = shipdate + rng.integers(1, 31, n)

    returned = receiptdate <= RECEIPT_CUTOFF
    ra = rng.integers(0, 2, n)                        # 0 -> 'R', 1 -> 'A'
    returnflag = np.where(returned, np.where(ra == 0, 2, 0), 1).astype(np.uint8)
    linestatus = (shipdate > RECEIPT_CUTOFF).astype(np.uint8)  # 1='O', 0='F'

    # spec 4.2.3: O_TOTALPRICE = sum over the order's lines of
    # extendedprice * (1 + tax) * (1 - discount)
    line_total = extendedprice * (1.0 + tax) * (1.0 - discount)
    starts = np.concatenate(([0], np.cumsum(lines_per_order)[:-1]))
    orders["o_totalprice"] = np.add.reduceat(line_total, starts)
    # spec 4.2.3: O_ORDERSTATUS = F if every line F, O if every line O,
    # else P (codes 0/1/2 into ["F", "O", "P"])
    f_cnt = np.add.reduceat((linestatus == 0).astype(np.int64), starts)
    status = np.full(len(starts), 2, dtype=np.uint8)
    status[f_cnt == lines_per_order] = 0
    status[f_cnt == 0] = 1
    orders["o_orderstatus"] = status

    # spec 4.2.3: L_SUPPKEY is one of the part's FOUR partsupp suppliers
    # (the same spread formula as gen_partsupp), picked uniformly — the
    # invariant Q9's ps join relies on
    S = n_suppliers(sf)
    i4 = rng.integers(0, 4, n)
    l_suppkey = ((partkey + i4 * (S // 4 + (partkey - 1) // S)) % S) + 1
    out = {
        "l_orderkey": l_orderkey,
        "l_suppkey": l_suppkey,
        "l_quantity": quantity,
        "l_extendedprice": extendedprice,
        "l_discount": discount,
        "l_tax": tax,
        "l_returnflag": returnflag,
        "l_linestatus": linestatus,
        "l_shipdate": shipdate.astype(np.int32),
        "l_commitdate": commitdate.astype(np.int32),
        "l_receiptdate": receiptdate.astype(np.int32),
        "l_partkey": partkey,
    }
    # drawn LAST so earlier columns' RNG streams stay fixture-stable
    out["l_shipmode"] = rng.integers(0, len(SHIPMODE), n).astype(np.uint8)
    out["l_shipinstruct"] = rng.integers(
        0, len(SHIPINSTRUCT), n).astype(np.uint8)
    return out


def gen_customer(sf, seed=42, strings=False):
    """customer columns: c_custkey i64 (dense 1..N), c_mktsegment u8 code,
    c_nationkey i32, c_acctbal f64 (spec U[-999.99, 9999.99]). With
    strings=True also c_name ('Customer#%09d', the spec format), c_phone
    (spec 4.2.2.9 country-code format) and placeholder c_address/
    c_comment (spec text grammar out of scope, tpch_gen header) — the
    Q10 output attributes, all functionally dependent on c_custkey."""
    n = n_customers(sf)
    rng = np.random.default_rng([seed, 3])
    out = {
        "c_custkey": np.arange(1, n + 1, dtype=np.int64),
        "c_mktsegment": rng.integers(0, len(MKTSEGMENT), n).astype(np.uint8),
        "c_nationkey": rng.integers(0, 25, n).astype(np.int32),
        "c_acctbal": rng.integers(-99999, 1000000, n) / 100.0,
    }
    if strings:
        ck = out["c_custkey"]
        nk = out["c_nationkey"]
        local = rng.integers(100, 1000, (n, 3))
        out["c_name"] = np.array(
            ["Customer#%09d" % k for k in ck], dtype=object)
        out["c_phone"] = np.array(
            ["%d-%d-%d-%d" % (10 + nk[i], local[i, 0], local[i, 1],
                              local[i, 2]) for i in range(n)], dtype=object)
        out["c_address"] = np.array(
            ["addr#%d.%d" % (k, int(rng.integers(0, 1 << 30)))
             for k in ck], dtype=object)
        out["c_comment"] = np.array(
            ["comment#%d" % k for k in ck], dtype=object)
    return out


def gen_supplier(sf, seed=42):
    n = n_suppliers(sf)
    rng = np.random.default_rng([seed, 4])
    out = {
        "s_suppkey": np.arange(1, n + 1, dtype=np.int64),
        "s_nationkey": rng.integers(0, 25, n).astype(np.int32),
    }
    # appended draws (stream-stable): spec s_acctbal U[-999.99, 9999.99];
    # s_comment LIKE '%Customer%Complaints%' as a flag (text boundary,
    # header; dbgen plants it in 10/SF/10000 suppliers)
    out["s_acctbal"] = rng.integers(-99999, 1000000, n) / 100.0
    out["s_comment_complaints"] = (rng.random(n) < 0.0013).astype(np.uint8)
    return out


def gen_part(sf, seed=42):
    """part columns: p_partkey i64 (dense 1..N — the key space
    gen_lineitem's l_partkey draws from), p_type u8 code into the 150
    spec type combinations (PTYPE_SYL* above); LIKE 'PROMO%' == code //
    25 == PTYPE_PROMO_SYL1."""
    n = n_parts(sf)
    rng = np.random.default_rng([seed, 5])
    out = {
        "p_partkey": np.arange(1, n + 1, dtype=np.int64),
        "p_type": rng.integers(0, 150, n).astype(np.uint8),
    }
    # Q17/Q19 attributes, drawn after p_type (stream-append stable):
    # p_brand 'Brand#MN' M,N in 1..5 -> code 0..24 (brand_code);
    # p_container 5x8 spec combinations -> code 0..39 (container_code);
    # p_size uniform 1..50 (spec 4.2.3)
    out["p_brand"] = rng.integers(0, 25, n).astype(np.uint8)
    out["p_container"] = rng.integers(0, 40, n).astype(np.uint8)
    out["p_size"] = rng.integers(1, 51, n).astype(np.uint8)
    # p_name's FIRST word as a code (spec: 5 of 92 color words; prefix
    # LIKE 'forest%' == first word == "forest" == code P_NAME_FOREST —
    # the text-boundary adaptation documented in the header)
    out["p_name1"] = rng.integers(0, 92, n).astype(np.uint8)
    # p_name LIKE '%green%' as a flag: 5 words drawn from 92, so
    # P(contains a given word) = 1 - C(91,5)/C(92,5) = 5/92
    out["p_name_green"] = (rng.random(n) < 5 / 92).astype(np.uint8)
    return out


def gen_partsupp(sf, seed=42):
    """partsupp: 4 rows per part (spec 4.2.3); ps_suppkey follows the
    spec's supplier-spread formula
        ((ps_partkey + i*(S/4 + (ps_partkey-1)/S)) % S) + 1,  i in 0..3
    (guarantees 4 distinct suppliers per part); ps_availqty U[1,9999];
    ps_supplycost U[1.00, 1000.00]."""
    npart = n_parts(sf)
    S = n_suppliers(sf)
    rng = np.random.default_rng([seed, 6])
    pk = np.repeat(np.arange(1, npart + 1, dtype=np.int64), 4)
    i = np.tile(np.arange(4, dtype=np.int64), npart)
    sk = ((pk + i * (S // 4 + (pk - 1) // S)) % S) + 1
    n = len(pk)
    return {
        "ps_partkey": pk,
        "ps_suppkey": sk,
        "ps_availqty": rng.integers(1, 10_000, n).astype(np.int32),
        "ps_supplycost": rng.integers(100, 100_001, n) / 100.0,
    }


def gen_nation():
    return {
        "n_nationkey": np.arange(25, dtype=np.int32),
        "n_regionkey": np.array([r for _, r in NATIONS], dtype=np.int32),
        "n_name": np.array([n for n, _ in NATIONS], dtype=object),
    }


def gen_region():
    return {"r_regionkey": np.arange(5, dtype=np.int32)}


def gen_all(sf, seed=42):
    orders = gen_orders(sf, seed)
    return {
        "orders": orders,
        "lineitem": gen_lineitem(sf, seed, orders),
        "customer": gen_customer(sf, seed),
        "supplier": gen_supplier(sf, seed),
        "part": gen_part(sf, seed),
        "partsupp": gen_partsupp(sf, seed),
        "nation": gen_nation(),
        "region": gen_region(),
    }
