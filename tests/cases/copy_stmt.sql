CREATE TABLE cp (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h));
INSERT INTO cp (h, ts, v) VALUES ('a',1,1.0),('b',2,2.0);
COPY cp TO '/tmp/gdb_golden_copy.parquet';
CREATE TABLE cp2 (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h));
COPY cp2 FROM '/tmp/gdb_golden_copy.parquet';
SELECT h, v FROM cp2 ORDER BY h
