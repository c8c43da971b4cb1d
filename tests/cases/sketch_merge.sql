CREATE TABLE sk2 (ts TIMESTAMP TIME INDEX, h STRING PRIMARY KEY, v DOUBLE);
INSERT INTO sk2 VALUES (1000,'a',1),(2000,'a',2),(3000,'b',2),(4000,'b',3),(5000,'c',9);
SELECT hll_count(hll(h)) FROM sk2;
SELECT uddsketch_calc(0.5, uddsketch_state(128, 0.01, v)) > 1.9 FROM sk2;
SELECT approx_percentile(v, 0.99) FROM sk2;
