"""Synthetic uniref50-shaped FASTA for training evidence (no network)."""
import random, sys
rng = random.Random(0)
AA = "ACDEFGHIKLMNPQRSTVWY"
n = int(sys.argv[1]) if len(sys.argv) > 1 else 20000
with open("synthetic.fasta", "w") as f:
    for i in range(n):
        L = rng.randint(80, 500)
        seq = "".join(rng.choice(AA) for _ in range(L))
        f.write(f">UniRef50_S{i:06d} Synthetic protein n=1 Tax=Escherichia coli TaxID=562 RepID=S{i}_ECOLI\n")
        for j in range(0, L, 60):
            f.write(seq[j:j+60] + "\n")
print("wrote synthetic.fasta", n)
