CREATE TABLE af (ts TIMESTAMP TIME INDEX, h STRING PRIMARY KEY, v DOUBLE);
INSERT INTO af VALUES (1000,'a',1),(2000,'a',2),(3000,'a',3),(4000,'a',4),(5000,'a',100);
SELECT approx_percentile(v, 0.5) FROM af;
SELECT median(v) FROM af;
SELECT h, approx_percentile(v, 0.8) FROM af GROUP BY h;
