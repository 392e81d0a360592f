import sqlite3, sys, collections, re
db, out = sys.argv[1], sys.argv[2]
c = sqlite3.connect(db)
tabs = [r[0] for r in c.execute("select name from sqlite_master where type in ('table','view')")]
cc = [t for t in tabs if 'counters_collection' in t]
cols = [r[1] for r in c.execute(f"pragma table_info({cc[0]})")]
with open(out, 'w') as f:
    f.write("schema: " + ",".join(cols) + "\n")
    # aggregate counter value per kernel name
    q = f"select kernel_name, counter_name, sum(value), count(*) from {cc[0]} group by kernel_name, counter_name"
    try:
        for kn, cn, v, n in c.execute(q):
            short = re.sub(r'\(.*', '', kn.replace('(anonymous namespace)::', ''))[:70]
            f.write(f"{short}\t{cn}\t{v}\t{n}\n")
    except Exception as e:
        f.write(f"fallback: {e}\n")
        for row in c.execute(f"select * from {cc[0]} limit 200"):
            f.write(str(row)[:200] + "\n")
print("done")
