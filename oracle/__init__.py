# ORACLE — TEST INFRASTRUCTURE ONLY.
# CPU restatement of the reference (tuplex/tuplex) TransformStage semantics, used as
# the parity checker. Only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline
# leg may import or execute anything under oracle/. The product (tuplex_amd) must
# never import this package; the product's GPU path fails loudly if its HIP extension
# is missing.
