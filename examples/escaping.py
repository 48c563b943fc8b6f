# Quoting/escaping torture test: every line must come back byte-for-byte
# on stdout (no shell interpolation, no quote mangling in transit).
print("single 'quotes' inside double")
print('double "quotes" inside single')
print("escaped \"double\" quotes")
print("newline:\nnext line")
print("tab:\tafter tab")
print("backslash: \\ and raw $DOLLAR ${BRACES} `backticks`")
