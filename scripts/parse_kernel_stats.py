import csv
import sys

for row in csv.DictReader(open(sys.argv[1])):
    n = row["Name"].split("(")[0]
    if len(sys.argv) < 3 or sys.argv[2] in n:
        print(f"{n[:40]:42s} x{row['Calls']:>4} avg {float(row['AverageNs'])/1e6:8.3f} ms  {row['Percentage'][:5]}%")
